"""Host-proxy lifecycle (reference: hostproxy/manager.go EnsureRunning —
detached subprocess, idempotent)."""
from __future__ import annotations

import http.client
import os
import signal
import socket
import subprocess
import sys
import time
from pathlib import Path

from .. import consts
from ..errors import ClawkerError
from .daemon import pid_path, sock_path


class _UnixHTTPConnection(http.client.HTTPConnection):
    def __init__(self, path: str, timeout: float = 10.0):
        super().__init__("localhost", timeout=timeout)
        self._path = path

    def connect(self):
        self.sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self.sock.settimeout(self.timeout)
        self.sock.connect(self._path)


class HostProxyManager:
    def running(self) -> bool:
        try:
            pid = int(pid_path().read_text())
            os.kill(pid, 0)
            return sock_path().exists()
        except (OSError, ValueError):
            return False

    def ensure_running(self, timeout: float = 5.0) -> None:
        if self.running() and self._healthy():
            return
        consts.log_dir().mkdir(parents=True, exist_ok=True)
        logf = open(consts.log_dir() / "hostproxy.out", "ab")
        subprocess.Popen(
            [sys.executable, "-m", "clawker_amd.hostproxy.daemon"],
            stdin=subprocess.DEVNULL, stdout=logf, stderr=logf,
            start_new_session=True,
            cwd=str(Path(__file__).resolve().parents[2]))
        logf.close()
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if self._healthy():
                return
            time.sleep(0.02)
        raise ClawkerError("hostproxy failed to start "
                           f"(see {consts.log_dir() / 'hostproxy.out'})")

    def stop(self) -> bool:
        if not self.running():
            return False
        try:
            os.kill(int(pid_path().read_text()), signal.SIGTERM)
            return True
        except (OSError, ValueError):
            return False

    def _healthy(self) -> bool:
        try:
            c = _UnixHTTPConnection(str(sock_path()), timeout=2)
            c.request("GET", "/healthz")
            ok = c.getresponse().status == 200
            c.close()
            return ok
        except OSError:
            return False

    def request(self, method: str, path: str, body: bytes = b"",
                content_type: str = "application/json") -> tuple[int, bytes]:
        c = _UnixHTTPConnection(str(sock_path()))
        headers = {"Content-Type": content_type,
                   "Content-Length": str(len(body))}
        c.request(method, path, body=body, headers=headers)
        r = c.getresponse()
        data = r.read()
        c.close()
        return r.status, data
