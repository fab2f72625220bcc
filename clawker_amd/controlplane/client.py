"""CP admin client: ensure-running + framed requests over the admin socket
(reference: controlplane/manager host-side lifecycle bootstrap.go:203
ensureRunning + controlplane/adminclient Dial)."""
from __future__ import annotations

import errno
import os
import socket
import subprocess
import sys
import time
from pathlib import Path

from .. import consts
from ..engine import wire
from ..errors import CPSOSError, ClawkerError
from ..logger import get as get_logger
from .daemon import admin_sock_path, pid_path

log = get_logger("cpclient")


class CPClient:
    def __init__(self, auto_start: bool = True, timeout: float = 10.0):
        self.auto_start = auto_start
        self.timeout = timeout
        self._sock: socket.socket | None = None

    # -- lifecycle -------------------------------------------------------------
    def running(self) -> bool:
        try:
            pid = int(pid_path().read_text())
            os.kill(pid, 0)
            return admin_sock_path().exists()
        except (OSError, ValueError):
            return False

    def ensure_running(self) -> None:
        """Idempotent daemon spawn + readiness poll (reference:
        ensureRunning + /healthz poll, bootstrap.go:184-203). Concurrent
        callers (8-way fleet/soak) serialize on a spawn flock so exactly
        one cpd ever starts; the daemon additionally refuses to start
        over a live sibling."""
        if self.running() and self._ping():
            return
        import fcntl
        rd = consts.runtime_dir()
        rd.mkdir(parents=True, exist_ok=True)
        with open(rd / "cpd.spawn.lock", "w") as lockf:
            fcntl.flock(lockf, fcntl.LOCK_EX)
            # a racer may have finished the spawn while we waited
            if self.running() and self._ping():
                return
            env = dict(os.environ)
            consts.log_dir().mkdir(parents=True, exist_ok=True)
            logf = open(consts.log_dir() / "cpd.out", "ab")
            subprocess.Popen(
                [sys.executable, "-m", "clawker_amd.controlplane.daemon"],
                stdin=subprocess.DEVNULL, stdout=logf, stderr=logf,
                start_new_session=True, env=env,
                cwd=str(Path(__file__).resolve().parents[2]))
            logf.close()
            deadline = time.monotonic() + self.timeout
            while time.monotonic() < deadline:
                if self._ping():
                    return
                time.sleep(0.02)
        raise CPSOSError("control plane failed to become ready",
                         assist=f"check {consts.log_dir() / 'cpd.out'}")

    def stop(self) -> bool:
        if not self.running():
            return False
        try:
            self.request({"op": "shutdown"})
        except ClawkerError:
            pass
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline:
            if not self.running():
                return True
            time.sleep(0.05)
        try:
            os.kill(int(pid_path().read_text()), 15)
        except (OSError, ValueError):
            pass
        return True

    # -- requests ---------------------------------------------------------------
    def _connect(self) -> socket.socket:
        # unix connect() returns EAGAIN instantly when the daemon's accept
        # queue is momentarily full (fleet cold-start storms) and
        # ECONNREFUSED in the bind→listen window of a daemon that is
        # coming up. Both are transient, but with very different budgets:
        # a saturated-but-alive daemon drains within seconds, while
        # ECONNREFUSED against a STALE socket file (daemon SIGKILLed) is
        # persistent — retrying it long makes every liveness ping of a
        # dead daemon hang, which stalled CP crash-recovery under load.
        start = time.monotonic()
        deadline_eagain = start + min(self.timeout, 5.0)
        deadline_refused = start + 0.3   # bind→listen gap is microseconds
        delay = 0.005
        while True:
            s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            s.settimeout(self.timeout)
            try:
                s.connect(str(admin_sock_path()))
                return s
            except OSError as e:
                s.close()
                now = time.monotonic()
                if e.errno == errno.EAGAIN and now < deadline_eagain:
                    pass
                elif (e.errno in (errno.ECONNREFUSED, errno.ECONNRESET)
                      and now < deadline_refused):
                    pass
                else:
                    raise
                time.sleep(delay)
                delay = min(delay * 2, 0.1)

    def _ping(self) -> bool:
        try:
            s = self._connect()
            wire.send_frame(s, {"op": "ping"})
            r = wire.recv_frame(s)
            s.close()
            return bool(r and r.get("ok") and r.get("ready"))
        except OSError:
            return False

    def request(self, req: dict) -> dict:
        if self.auto_start:
            self.ensure_running()
        try:
            s = self._connect()
            wire.send_frame(s, req)
            resp = wire.recv_frame(s)
            s.close()
        except OSError as e:
            raise ClawkerError(f"control plane unreachable: {e}") from e
        if resp is None:
            raise ClawkerError("control plane closed the connection")
        if not resp.get("ok"):
            raise ClawkerError(f"control plane error: {resp.get('error')}")
        return resp

    # -- typed ops ---------------------------------------------------------------
    def status(self) -> dict:
        return self.request({"op": "status"})

    def agents(self) -> list[dict]:
        return self.request({"op": "agents"})["agents"]

    def reload_policy(self) -> int:
        return int(self.request({"op": "reload_policy"})["sandboxes"])

    def bypass(self, seconds: int) -> None:
        self.request({"op": "bypass", "seconds": seconds})

    def events(self, n: int = 100) -> list[dict]:
        return self.request({"op": "events", "n": n})["events"]

    def follow_events(self):
        """Generator of live events pushed by cpd's pub/sub (no polling).
        Ends when the daemon goes away; caller breaks to disconnect."""
        if self.auto_start:
            self.ensure_running()
        s = self._connect()
        try:
            wire.send_frame(s, {"op": "events_follow"})
            hello = wire.recv_frame(s)
            # a quiet period must not kill the stream: the timeout guards
            # connect/handshake only (cpd going away closes the socket)
            s.settimeout(None)
            if not (hello and hello.get("ok")):
                raise ClawkerError(
                    f"control plane error: {(hello or {}).get('error')}")
            while True:
                frame = wire.recv_frame(s)
                if frame is None:
                    return
                yield frame["event"]
        finally:
            s.close()
