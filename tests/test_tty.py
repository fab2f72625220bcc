"""Interactive TTY tests (reference hard-part: terminal fidelity —
docker/pty.go raw mode, resize, attach ordering)."""
import os
import pty
import select
import subprocess
import sys
import time
from pathlib import Path

import pytest

from conftest import requires_isolation

pytestmark = requires_isolation

REPO = Path(__file__).resolve().parent.parent


def _drain(fd, until: bytes, timeout=20.0) -> bytes:
    buf = b""
    end = time.monotonic() + timeout
    while time.monotonic() < end:
        r, _, _ = select.select([fd], [], [], 0.2)
        if r:
            try:
                chunk = os.read(fd, 4096)
            except OSError:
                break
            if not chunk:
                break
            buf += chunk
            if until in buf:
                return buf
    return buf


@pytest.fixture
def proj(isolated_env, tmp_path):
    root = tmp_path / "ttyproj"
    root.mkdir()
    (root / ".clawker.yaml").write_text("project: ttytest\nagent:\n  harness: echo\n")
    return root


def test_interactive_shell_roundtrip(proj):
    """clawker run -it with a real PTY: the in-sandbox sh sees a tty,
    echoes input, and the exit code propagates."""
    master, slave = pty.openpty()
    env = dict(os.environ, PYTHONPATH=str(REPO), TERM="xterm")
    p = subprocess.Popen(
        [sys.executable, "-m", "clawker_amd", "run", "--rm", "-i", "-t",
         "--no-firewall", "--no-host-services", "--",
         "/bin/sh", "-c", "echo TTY=$(tty); read x; echo GOT=$x; exit 4"],
        stdin=slave, stdout=slave, stderr=slave, env=env, cwd=str(proj),
        close_fds=True)
    os.close(slave)
    try:
        out = _drain(master, b"TTY=")
        assert b"TTY=/dev/pts/" in out, out      # real pty inside the sandbox
        os.write(master, b"hello-tty\r")
        out += _drain(master, b"GOT=hello-tty")
        assert b"GOT=hello-tty" in out, out
        p.wait(timeout=20)
        assert p.returncode == 4
    finally:
        os.close(master)
        if p.poll() is None:
            p.kill()


def test_tty_window_size_propagates(proj):
    """SIGWINCH-driven resize frames reach the sandbox pty."""
    master, slave = pty.openpty()
    import fcntl
    import struct
    import termios
    # set an unusual host terminal size before starting
    fcntl.ioctl(slave, termios.TIOCSWINSZ, struct.pack("HHHH", 37, 91, 0, 0))
    env = dict(os.environ, PYTHONPATH=str(REPO), TERM="xterm")
    p = subprocess.Popen(
        [sys.executable, "-m", "clawker_amd", "run", "--rm", "-i", "-t",
         "--no-firewall", "--no-host-services", "--",
         "/bin/sh", "-c", "sleep 0.3; stty size"],
        stdin=slave, stdout=slave, stderr=slave, env=env, cwd=str(proj),
        close_fds=True)
    os.close(slave)
    try:
        out = _drain(master, b"37 91")
        assert b"37 91" in out, out
        p.wait(timeout=20)
    finally:
        os.close(master)
        if p.poll() is None:
            p.kill()


def test_interactive_exec_tty(proj):
    """clawker exec -it: real pty inside the exec, input roundtrip, exit
    code propagation."""
    master, slave = pty.openpty()
    env = dict(os.environ, PYTHONPATH=str(REPO), TERM="xterm")
    # background sandbox to exec into
    up = subprocess.run(
        [sys.executable, "-m", "clawker_amd", "run", "-d", "--agent", "xt",
         "--no-firewall", "--no-host-services", "--", "sleep", "60"],
        capture_output=True, text=True, timeout=180, cwd=str(proj), env=env)
    assert up.returncode == 0, up.stderr
    p = subprocess.Popen(
        [sys.executable, "-m", "clawker_amd", "exec", "-i", "-t", "xt", "--",
         "/bin/sh", "-c", "echo XTTY=$(tty); read v; echo XGOT=$v; exit 6"],
        stdin=slave, stdout=slave, stderr=slave, env=env, cwd=str(proj),
        close_fds=True)
    os.close(slave)
    try:
        out = _drain(master, b"XTTY=")
        assert b"XTTY=/dev/pts/" in out, out
        os.write(master, b"ping\r")
        out += _drain(master, b"XGOT=ping")
        assert b"XGOT=ping" in out, out
        p.wait(timeout=20)
        assert p.returncode == 6
    finally:
        os.close(master)
        if p.poll() is None:
            p.kill()
        subprocess.run([sys.executable, "-m", "clawker_amd", "rm", "-f", "xt"],
                       capture_output=True, cwd=str(proj), env=env)


def test_iostreams_markdown_and_page_fallback():
    """Markdown render + pager fallback on non-tty (reference:
    iostreams pager/markdown surface)."""
    from clawker_amd.iostreams import TestIOStreams
    io_ = TestIOStreams()
    io_.markdown("# Title\n\n- item **bold**\n")
    assert "Title" in io_.out
    assert "item" in io_.out
    io2 = TestIOStreams()
    io2.page("long output\n" * 5)      # non-tty: plain print, no pager
    assert io2.out.count("long output") == 5
    io3 = TestIOStreams()
    io3.page("# H\n\ntext", markdown=True)
    assert "text" in io3.out


def test_iostreams_alt_screen_noop_on_non_tty():
    from clawker_amd.iostreams import TestIOStreams
    io_ = TestIOStreams()
    with io_.alt_screen():
        io_.print("inside")
    assert "\x1b[?1049h" not in io_.out     # no escape codes off-tty
    assert "inside" in io_.out
