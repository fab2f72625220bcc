"""Client for the ckd control socket (the reference's ClawkerdService
Session stream, api/clawkerd/v1/clawkerd.proto — reimagined as framed JSON
over a Unix socket; see native/ckd/ckd.cpp for the protocol)."""
from __future__ import annotations

import socket
import time
from pathlib import Path
from typing import Any, Callable, Iterator

from ..errors import EngineError
from . import wire


class CkdClient:
    def __init__(self, sock_path: Path, timeout: float | None = 30.0):
        self.sock_path = sock_path
        try:
            self.sock = wire.connect_unix(sock_path, timeout=timeout)
        except OSError as e:
            raise EngineError("ckd connect", f"{sock_path}: {e}") from e

    @classmethod
    def wait_connect(cls, sock_path: Path, deadline_s: float = 10.0,
                     timeout: float | None = 30.0) -> "CkdClient":
        """Connect, retrying until ckd has bound its socket (sandbox boot)."""
        end = time.monotonic() + deadline_s
        last: Exception | None = None
        # exponential backoff from 0.3 ms: ckd binds ~1-3 ms after the
        # shim forks, so a fixed 5 ms poll quantized the whole cold
        # start (headline metric) to its own granularity
        delay = 0.0003
        while time.monotonic() < end:
            try:
                return cls(sock_path, timeout=timeout)
            except EngineError as e:
                last = e
                time.sleep(delay)
                delay = min(delay * 2, 0.005)
        raise EngineError("ckd connect", f"timed out after {deadline_s}s: {last}")

    def close(self) -> None:
        try:
            self.sock.close()
        except OSError:
            pass

    def __enter__(self) -> "CkdClient":
        return self

    def __exit__(self, *exc: Any) -> None:
        self.close()

    # -- basic request/response ----------------------------------------------
    def send(self, obj: dict[str, Any]) -> None:
        wire.send_frame(self.sock, obj)

    def recv(self) -> dict[str, Any] | None:
        return wire.recv_frame(self.sock)

    def hello(self) -> dict[str, Any]:
        self.send({"t": "hello"})
        r = self.recv()
        if r is None or r.get("t") != "hello":
            raise EngineError("ckd hello", f"unexpected reply: {r}")
        return r

    def agent_ready(self, cmd: list[str] | None = None) -> int:
        """Release the agent CMD (reference: boot_steps.go AgentReady).
        Returns the agent pid."""
        msg: dict[str, Any] = {"t": "agent_ready"}
        if cmd:
            msg["cmd"] = cmd
        self.send(msg)
        r = self._wait_for("ready_ack")
        return int(r.get("pid", -1))

    def agent_initialized(self) -> None:
        self.send({"t": "agent_initialized"})
        self._wait_for("ok")

    def status(self) -> dict[str, Any]:
        self.send({"t": "status"})
        return self._wait_for("status")

    def signal(self, sig: int) -> None:
        self.send({"t": "signal", "sig": sig})

    def resize(self, rows: int, cols: int) -> None:
        self.send({"t": "resize", "rows": rows, "cols": cols})

    def write_stdin(self, data: bytes) -> None:
        self.send({"t": "stdin", "data": wire.b64(data)})

    def close_stdin(self) -> None:
        self.send({"t": "close_stdin"})

    def attach(self) -> None:
        """Subscribe this connection to the console stream."""
        self.send({"t": "attach"})
        self._wait_for("attached")

    def _wait_for(self, t: str) -> dict[str, Any]:
        while True:
            r = self.recv()
            if r is None:
                raise EngineError("ckd", "connection closed")
            if r.get("t") == "error":
                raise EngineError("ckd", r.get("msg", "error"))
            if r.get("t") == t:
                return r
            # skip interleaved events (console/agent_exit broadcasts)

    # -- exec ----------------------------------------------------------------
    _exec_seq = 0

    def exec(
        self,
        stages: list[dict[str, Any]],
        stdin: bytes = b"",
        env: dict[str, str] | None = None,
        on_output: Callable[[int, bytes], None] | None = None,
    ) -> tuple[int, bytes, bytes]:
        """Run a staged pipeline inside the sandbox; each stage may carry
        uid/gid/user/cwd (reference: ShellCommand PipeStage semantics,
        clawkerd.proto:90-122). Returns (last_stage_code, stdout, stderr)."""
        CkdClient._exec_seq += 1
        eid = f"x{CkdClient._exec_seq}-{int(time.time() * 1000) & 0xFFFFFF}"
        msg: dict[str, Any] = {"t": "exec", "id": eid, "stages": stages}
        if stdin:
            msg["stdin"] = wire.b64(stdin)
        if env:
            msg["env"] = env
        self.send(msg)
        out = bytearray()
        err = bytearray()
        code = -1
        # the socket timeout guards connect/handshake, NOT command
        # duration: a 30s recv timeout would kill any long-running exec
        # (agents run long commands constantly). ckd dying still ends the
        # stream: its socket closes and recv() returns None.
        prev_to = self.sock.gettimeout()
        self.sock.settimeout(None)
        try:
            while True:
                r = self.recv()
                if r is None:
                    raise EngineError("ckd exec", "connection closed mid-exec")
                t = r.get("t")
                if t == "out" and r.get("id") == eid:
                    data = wire.unb64(r.get("data", ""))
                    if r.get("stream") == 1:
                        out.extend(data)
                    else:
                        err.extend(data)
                    if on_output:
                        on_output(int(r.get("stream", 1)), data)
                elif t == "done" and r.get("id") == eid:
                    code = int(r.get("code", -1))
                    break
                elif t == "error":
                    raise EngineError("ckd exec", r.get("msg", "error"))
                # ignore unrelated events
        finally:
            self.sock.settimeout(prev_to)
        return code, bytes(out), bytes(err)

    def exec_start(self, stages: list[dict[str, Any]], stdin: bytes = b"",
                   env: dict[str, str] | None = None) -> str:
        """Fire-and-forget exec (docker exec -d): returns the exec id once
        ckd has spawned it. The job keeps running after this client
        disconnects (ckd orphans it; output goes to the void)."""
        CkdClient._exec_seq += 1
        eid = f"d{CkdClient._exec_seq}-{int(time.time() * 1000) & 0xFFFFFF}"
        msg: dict[str, Any] = {"t": "exec", "id": eid, "stages": stages}
        if stdin:
            msg["stdin"] = wire.b64(stdin)
        if env:
            msg["env"] = env
        self.send(msg)
        self._wait_for("started")
        return eid

    def exec_start_tty(self, argv: list[str], user: str = "", cwd: str = "",
                       env: dict[str, str] | None = None) -> str:
        """Start an interactive (pty) exec; returns its id. The caller
        pumps frames: out(id) frames arrive, exec_stdin/exec_resize go
        back, done(id) carries the exit code."""
        CkdClient._exec_seq += 1
        eid = f"t{CkdClient._exec_seq}-{int(time.time() * 1000) & 0xFFFFFF}"
        stage: dict[str, Any] = {"argv": argv}
        if user:
            stage["user"] = user
        if cwd:
            stage["cwd"] = cwd
        msg: dict[str, Any] = {"t": "exec", "id": eid, "tty": True,
                               "stages": [stage]}
        if env:
            msg["env"] = env
        self.send(msg)
        self._wait_for("started")
        return eid

    def exec_stdin(self, eid: str, data: bytes) -> None:
        self.send({"t": "exec_stdin", "id": eid, "data": wire.b64(data)})

    def exec_resize(self, eid: str, rows: int, cols: int) -> None:
        self.send({"t": "exec_resize", "id": eid, "rows": rows, "cols": cols})

    # -- console streaming ---------------------------------------------------
    def stream_events(self) -> Iterator[dict[str, Any]]:
        """Yield frames until the connection closes (use after attach())."""
        while True:
            r = self.recv()
            if r is None:
                return
            yield r
