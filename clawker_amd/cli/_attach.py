"""Terminal attach: pump the host terminal <-> ckd console stream.

Reference: internal/docker/pty.go PTYHandler (raw mode, alt-screen-aware
restore, resize propagation) + run.go attach goroutines. Here the console
travels as frames over the ckd control socket; stdin is forwarded as
stdin frames; SIGWINCH becomes resize frames.
"""
from __future__ import annotations

import os
import select
import signal
import sys
import termios
import tty

from ..engine import CkdClient
from ..engine import wire


def _term_size(fd: int) -> tuple[int, int]:
    try:
        sz = os.get_terminal_size(fd)
        return sz.lines, sz.columns
    except OSError:
        return 24, 80


def stream(client: CkdClient, interactive: bool, tty_mode: bool) -> int:
    """Attach to a sandbox console; returns the agent exit code (or -1 if
    the stream ended without an agent_exit event)."""
    stdin_fd = sys.stdin.fileno() if interactive else -1
    stdout = sys.stdout.buffer
    exit_code = -1
    saved = None
    if tty_mode and interactive and sys.stdin.isatty():
        saved = termios.tcgetattr(stdin_fd)
        tty.setraw(stdin_fd)
        rows, cols = _term_size(sys.stdout.fileno())
        client.resize(rows, cols)

        def on_winch(*_a):
            r, c = _term_size(sys.stdout.fileno())
            try:
                client.resize(r, c)
            except OSError:
                pass

        signal.signal(signal.SIGWINCH, on_winch)

    sock = client.sock
    sock.settimeout(None)
    try:
        while True:
            rlist = [sock]
            if stdin_fd >= 0:
                rlist.append(stdin_fd)
            try:
                ready, _, _ = select.select(rlist, [], [])
            except InterruptedError:
                continue
            if stdin_fd in ready:
                data = os.read(stdin_fd, 65536)
                if not data:
                    client.close_stdin()
                    stdin_fd = -1
                else:
                    if tty_mode and data == b"\x10\x11":   # ctrl-p ctrl-q detach
                        return -2
                    client.write_stdin(data)
            if sock in ready:
                frame = wire.recv_frame(sock)
                if frame is None:
                    break
                t = frame.get("t")
                if t == "console":
                    stdout.write(wire.unb64(frame.get("data", "")))
                    stdout.flush()
                elif t == "agent_exit":
                    exit_code = int(frame.get("code", -1))
                    break
    finally:
        if saved is not None:
            termios.tcsetattr(stdin_fd if stdin_fd >= 0 else sys.stdin.fileno(),
                              termios.TCSADRAIN, saved)
            signal.signal(signal.SIGWINCH, signal.SIG_DFL)
    return exit_code
