"""hostproxyd — host services for in-sandbox agents.

Reference: internal/hostproxy (HTTP daemon :18374; routes server.go:99-109:
POST /open/url — open in the host browser, POST /git/credential — fill
from the host git credential store, POST /callback/register +
GET /cb/{session}/... — OAuth callback interception with TTL sessions).

Single-node redesign: serves over a Unix socket in the runtime dir
(bind-mounted into every sandbox at /run/clawker/hostproxy.sock) plus a
loopback HTTP port for OAuth callbacks (browsers redirect to
http://127.0.0.1:18374/cb/...). The proxy is itself an attack surface, so
/open/url re-checks the egress rules store exactly like the reference's
egress_check.go mirror — fail-closed if the rules file is unreadable.
Self-exits when no sandboxes remain (reference: 0-containers + grace).
"""
from __future__ import annotations

import json
import os
import secrets
import shutil
import signal
import socket
import subprocess
import threading
import time
import urllib.parse
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from pathlib import Path

from .. import consts
from ..logger import get as get_logger, setup as logger_setup

log = get_logger("hostproxy")

SESSION_TTL_S = 600


def sock_path() -> Path:
    return consts.runtime_dir() / consts.HOSTPROXY_SOCK


def pid_path() -> Path:
    return consts.runtime_dir() / "hostproxy.pid"


class CallbackStore:
    """OAuth callback sessions with TTL (reference: SessionStore)."""

    def __init__(self):
        self._sessions: dict[str, dict] = {}
        self._lock = threading.Lock()

    MAX_SESSIONS = 256
    MAX_HITS = 64

    def register(self, target_port: int, sandbox: str) -> str:
        sid = secrets.token_urlsafe(16)
        with self._lock:
            self._gc()
            if len(self._sessions) >= self.MAX_SESSIONS:
                # bounded: a misbehaving agent cannot grow host memory
                oldest = min(self._sessions, key=lambda k:
                             self._sessions[k]["created"])
                del self._sessions[oldest]
            self._sessions[sid] = {
                "port": target_port, "sandbox": sandbox,
                "created": time.time(), "hits": []}
        return sid

    def get(self, sid: str) -> dict | None:
        with self._lock:
            self._gc()
            return self._sessions.get(sid)

    def _gc(self) -> None:
        cutoff = time.time() - SESSION_TTL_S
        for sid in [s for s, v in self._sessions.items() if v["created"] < cutoff]:
            del self._sessions[sid]


class HostProxyHandlerBase(BaseHTTPRequestHandler):
    callbacks: CallbackStore
    protocol_version = "HTTP/1.1"

    # -- helpers -------------------------------------------------------------
    def _json(self, code: int, obj: dict) -> None:
        body = json.dumps(obj).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def _body(self) -> bytes:
        n = int(self.headers.get("Content-Length") or 0)
        return self.rfile.read(n) if n else b""

    def log_message(self, *a):
        pass

    # -- routes --------------------------------------------------------------
    def do_POST(self):   # noqa: N802
        if self.path == "/open/url":
            self._open_url()
        elif self.path == "/git/credential":
            self._git_credential()
        elif self.path == "/callback/register":
            self._callback_register()
        else:
            self._json(404, {"error": "not found"})

    def do_GET(self):   # noqa: N802
        if self.path.startswith("/cb/"):
            self._callback_hit()
        elif self.path.startswith("/callback/poll/"):
            self._callback_poll()
        elif self.path == "/healthz":
            self._json(200, {"ok": True})
        else:
            self._json(404, {"error": "not found"})

    def _callback_poll(self) -> None:
        """Agents poll for captured OAuth redirects (completing the loop:
        register -> user browser hits /cb/<sid>/... -> agent polls)."""
        sid = self.path.rsplit("/", 1)[1]
        sess = self.callbacks.get(sid)
        if sess is None:
            self._json(404, {"error": "unknown callback session"})
            return
        self._json(200, {"hits": sess["hits"]})

    def _open_url(self) -> None:
        """Open a URL in the host browser — but only if the egress policy
        would allow the domain (reference: egress_check.go:17-60 mirrored
        rule semantics, fail-closed)."""
        try:
            req = json.loads(self._body() or b"{}")
            url = req.get("url", "")
            parsed = urllib.parse.urlsplit(url)
            if parsed.scheme not in ("http", "https") or not parsed.hostname:
                self._json(400, {"error": "bad url"})
                return
            try:
                from ..firewall import EgressRulesStore
                store = EgressRulesStore()
                proto = "tls" if parsed.scheme == "https" else "http"
                port = parsed.port or (443 if proto == "tls" else 80)
                rule = store.match_domain(parsed.hostname, proto, port)
                if rule is None:
                    # domain-only fallback: any rule for the host
                    import fnmatch
                    rule = next(
                        (r for r in store.list()
                         if r.dst.lower() == parsed.hostname.lower()
                         or fnmatch.fnmatch(parsed.hostname.lower(), r.dst.lower())),
                        None)
                if rule is None:
                    log.warn("open_url_denied", url=url)
                    self._json(403, {"error": "url not in egress policy"})
                    return
                if not store.path_allowed(rule, parsed.path or "/"):
                    self._json(403, {"error": "path denied by egress policy"})
                    return
            except Exception as e:
                # FAIL CLOSED on policy errors
                log.error("open_url_policy_error", err=str(e))
                self._json(403, {"error": "policy unavailable (fail closed)"})
                return
            opener = shutil.which("xdg-open") or shutil.which("open")
            if opener:
                subprocess.Popen([opener, url], stdout=subprocess.DEVNULL,
                                 stderr=subprocess.DEVNULL)
                self._json(200, {"opened": True})
            else:
                self._json(200, {"opened": False, "url": url,
                                 "hint": "no host browser; open manually"})
        except (ValueError, OSError) as e:
            self._json(400, {"error": str(e)})

    def _git_credential(self) -> None:
        """`git credential fill` against the HOST credential store —
        credentials are never copied into sandboxes (reference:
        containerfs.go:1-12 doctrine + hostproxy git route). Requires a
        valid agent bootstrap token (the per-agent cert analog)."""
        from ..auth import verify_agent_token
        token = self.headers.get("X-Clawker-Token", "").strip()
        if not token or verify_agent_token(token) is None:
            log.warn("git_credential_denied", reason="bad agent token")
            self._json(403, {"error": "invalid agent token"})
            return
        try:
            payload = self._body().decode()
            r = subprocess.run(["git", "credential", "fill"], input=payload,
                               capture_output=True, text=True, timeout=20)
            if r.returncode != 0:
                self._json(502, {"error": r.stderr.strip()})
                return
            self.send_response(200)
            body = r.stdout.encode()
            self.send_header("Content-Type", "text/plain")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)
        except (OSError, subprocess.TimeoutExpired) as e:
            self._json(502, {"error": str(e)})

    def _callback_register(self) -> None:
        try:
            req = json.loads(self._body() or b"{}")
            sid = self.callbacks.register(int(req.get("port", 0)),
                                          req.get("sandbox", ""))
            self._json(200, {
                "session": sid,
                "callback_url": f"http://127.0.0.1:{consts.HOSTPROXY_PORT}/cb/{sid}/",
            })
        except (ValueError, KeyError) as e:
            self._json(400, {"error": str(e)})

    def _callback_hit(self) -> None:
        parts = self.path.split("/", 3)
        sid = parts[2] if len(parts) > 2 else ""
        sess = self.callbacks.get(sid)
        if sess is None:
            self._json(404, {"error": "unknown callback session"})
            return
        tail = "/" + (parts[3] if len(parts) > 3 else "")
        if len(sess["hits"]) < CallbackStore.MAX_HITS:
            sess["hits"].append({"path": tail, "ts": time.time()})
        # forward into the sandbox's loopback listener via its gateway?
        # single-node: the agent polls /callback/poll — respond OK here.
        body = (b"<html><body><h3>clawker: authentication complete.</h3>"
                b"You can close this tab.</body></html>")
        self.send_response(200)
        self.send_header("Content-Type", "text/html")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)


class UnixHTTPServer(ThreadingHTTPServer):
    address_family = socket.AF_UNIX

    def server_bind(self):
        Path(self.server_address).parent.mkdir(parents=True, exist_ok=True)
        try:
            os.unlink(self.server_address)
        except OSError:
            pass
        self.socket.bind(self.server_address)
        os.chmod(self.server_address, 0o666)

    def get_request(self):
        sock, _ = self.socket.accept()
        return sock, ("unix", 0)


def serve() -> int:
    callbacks = CallbackStore()
    handler = type("Handler", (HostProxyHandlerBase,), {"callbacks": callbacks})

    # pid BEFORE the socket becomes healthy: ensure_running returns on
    # health, and a caller checking running() (pid + socket) in that
    # instant must not see a half-registered daemon
    pid_path().parent.mkdir(parents=True, exist_ok=True)
    pid_path().write_text(str(os.getpid()))
    unix_srv = UnixHTTPServer(str(sock_path()), handler)
    threading.Thread(target=unix_srv.serve_forever, daemon=True).start()
    tcp_srv = None
    try:
        tcp_srv = ThreadingHTTPServer(("127.0.0.1", consts.HOSTPROXY_PORT), handler)
        threading.Thread(target=tcp_srv.serve_forever, daemon=True).start()
    except OSError as e:
        log.warn("hostproxy_tcp_unavailable", err=str(e))

    log.info("hostproxy_ready", sock=str(sock_path()))
    stop = threading.Event()
    signal.signal(signal.SIGTERM, lambda *a: stop.set())

    idle_since: float | None = None
    while not stop.is_set():
        stop.wait(2.0)
        # self-exit when no sandboxes remain (reference: 0 containers + 60s)
        n = 0
        try:
            from ..engine.state import StateDB
            db = StateDB()
            n = len(db.list_sandboxes())
            db.close()
        except Exception:
            # an erroring/deleted state dir counts as idle — otherwise a
            # test-spawned daemon lives forever (same bug class as the
            # cpd drain fix)
            n = 0
        if n == 0:
            idle_since = idle_since or time.time()
            if time.time() - idle_since > 60:
                log.info("hostproxy_idle_exit")
                break
        else:
            idle_since = None
    unix_srv.shutdown()
    if tcp_srv:
        tcp_srv.shutdown()
    sock_path().unlink(missing_ok=True)
    pid_path().unlink(missing_ok=True)
    return 0


def main() -> int:
    logger_setup(consts.log_dir() / "hostproxy.log")
    return serve()


if __name__ == "__main__":
    raise SystemExit(main())
