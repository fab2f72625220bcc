"""TLS MITM path-rule enforcement (reference: Envoy MITM filter chains
with the clawker CA — path rules on HTTPS destinations)."""
import json
import os
import socket
import ssl
import subprocess
import threading
import time
from pathlib import Path

import pytest

from conftest import requires_isolation


def _make_tls_upstream(tmp_path):
    """Self-signed HTTPS upstream on host loopback."""
    key = tmp_path / "up.key"
    crt = tmp_path / "up.crt"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "ec", "-pkeyopt",
         "ec_paramgen_curve:P-256", "-keyout", str(key), "-out", str(crt),
         "-nodes", "-subj", "/CN=secure.test", "-days", "2"],
        check=True, capture_output=True)
    import http.server

    class H(http.server.BaseHTTPRequestHandler):
        def do_GET(self):
            body = f"TLS_UPSTREAM path={self.path}".encode()
            self.send_response(200)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, *a):
            pass

    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), H)
    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
    ctx.load_cert_chain(str(crt), str(key))
    srv.socket = ctx.wrap_socket(srv.socket, server_side=True)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    return srv, srv.server_address[1]


def test_mitm_ca_and_leaf_minting(isolated_env):
    from clawker_amd.firewall import mitm
    crt, key = mitm.ensure_ca()
    crt2, _ = mitm.ensure_ca()
    assert crt == crt2 and crt.is_file()
    leaf_crt, leaf_key = mitm.leaf_for("api.example.com")
    # leaf verifies against the CA
    r = subprocess.run(["openssl", "verify", "-CAfile", str(crt), str(leaf_crt)],
                       capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    out = subprocess.run(["openssl", "x509", "-in", str(leaf_crt), "-noout",
                          "-text"], capture_output=True, text=True).stdout
    assert "DNS:api.example.com" in out
    # rotation drops leaves and reissues the CA
    mitm.rotate_ca()
    crt3, _ = mitm.ensure_ca()
    assert crt3.read_bytes() != b"" and not leaf_crt.exists()
    bundle = mitm.combined_trust_bundle()
    assert "BEGIN CERTIFICATE" in bundle.read_text()


def test_mitm_gateway_enforces_https_paths(isolated_env, tmp_path, monkeypatch):
    """Host-side check of the MITM chain via the gateway socket directly:
    CONNECT to a path-scoped TLS rule terminates TLS with the minted leaf,
    allows /api/*, rejects /share/*, keeps the session usable."""
    monkeypatch.setenv("CLAWKER_MITM_INSECURE_UPSTREAM", "1")
    srv, port = _make_tls_upstream(tmp_path)
    monkeypatch.setenv("CLAWKER_DNS_STATIC", "secure.test=127.0.0.1")
    from clawker_amd.config.schema import EgressRule
    from clawker_amd.firewall import EgressRulesStore, mitm
    from clawker_amd.firewall.gateway import GatewayManager
    from clawker_amd.firewall.policy import compile_policy, write_policy_snapshot
    EgressRulesStore().add([EgressRule(
        dst="secure.test", proto="tls", port=port,
        paths=["/api/"], deny_paths=["/share"])])
    rundir = tmp_path / "rundir"
    rundir.mkdir()
    write_policy_snapshot(rundir, compile_policy())
    events = []
    mgr = GatewayManager(on_event=events.append)
    mgr.attach("clawker.t.mitm", rundir)
    try:
        # speak proxy protocol to egress.sock like ckgw would relay
        raw = socket.socket(socket.AF_UNIX)
        raw.settimeout(10)
        raw.connect(str(rundir / "egress.sock"))
        raw.sendall(f"CONNECT secure.test:{port} HTTP/1.1\r\n\r\n".encode())
        assert b"200 Connection established" in raw.recv(100)
        # client TLS trusting OUR CA — proves the MITM leaf is presented
        ca_crt, _ = mitm.ensure_ca()
        cctx = ssl.create_default_context(cafile=str(ca_crt))
        tls = cctx.wrap_socket(raw, server_hostname="secure.test")
        leaf = tls.getpeercert()
        assert ("commonName", "secure.test") in leaf["subject"][0]

        def req(path):
            tls.sendall(f"GET {path} HTTP/1.1\r\nHost: secure.test\r\n\r\n".encode())
            data = b""
            while b"\r\n\r\n" not in data:
                chunk = tls.recv(65536)
                assert chunk, f"gateway closed mid-response (got {data!r})"
                data += chunk
            head, _, body = data.partition(b"\r\n\r\n")
            cl = [l for l in head.split(b"\r\n") if l.lower().startswith(b"content-length")]
            want = int(cl[0].split(b":")[1]) if cl else 0
            while len(body) < want:
                chunk = tls.recv(65536)
                assert chunk, "gateway closed mid-body"
                body += chunk
            return head.split(b" ")[1], body

        status, body = req("/api/v1/data")
        assert status == b"200" and b"TLS_UPSTREAM path=/api/v1/data" in body
        status, _ = req("/share/secret")
        assert status == b"403"
        status, body = req("/api/again")     # session survives the 403
        assert status == b"200" and b"/api/again" in body
        status, _ = req("/outside")          # not in the allow-list
        assert status == b"403"
        tls.close()
        kinds = {(e.get("action"), e.get("path")) for e in events if e.get("mitm")}
        assert ("allow", "/api/v1/data") in kinds
        assert ("deny", "/share/secret") in kinds
    finally:
        mgr.detach_all()
        srv.shutdown()


@requires_isolation
def test_mitm_end_to_end_from_sandbox(isolated_env, tmp_path, monkeypatch):
    """Full chain: in-sandbox python HTTPS client -> ckgw -> gateway MITM
    -> TLS upstream, trusting the staged bundle via SSL_CERT_FILE."""
    monkeypatch.setenv("CLAWKER_MITM_INSECURE_UPSTREAM", "1")
    srv, port = _make_tls_upstream(tmp_path)
    monkeypatch.setenv("CLAWKER_DNS_STATIC", "secure.test=127.0.0.1")
    ws = tmp_path / "mproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: mitmtest\n")
    from clawker_amd.config import load_config
    from clawker_amd.config.schema import EgressRule
    from clawker_amd.firewall import EgressRulesStore
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    EgressRulesStore().add([EgressRule(
        dst="secure.test", proto="tls", port=port, deny_paths=["/share"])])
    orch = Orchestrator(load_config(ws))
    name = "clawker.mitmtest.a"
    script = f"""
import json, urllib.request, urllib.error
out = {{}}
def get(url):
    try:
        with urllib.request.urlopen(url, timeout=10) as r:
            return r.status, r.read().decode()
    except urllib.error.HTTPError as e:
        return e.code, ""
    except Exception as e:
        return -1, str(e)
out["ok"] = get("https://secure.test:{port}/fine")
out["denied"] = get("https://secure.test:{port}/share/x")
print("MITM " + json.dumps(out), flush=True)
"""
    try:
        orch.run(RunOptions(agent="a", name=name, autostart=False, firewall=True,
                            cmd=["python3", "-c", script]))
        rundir = orch.engine.inspect(name).rundir
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline and not (rundir / "egress.sock").exists():
            time.sleep(0.05)
        assert (rundir / "trust-bundle.crt").exists()
        with orch.client(name) as c:
            c.agent_ready()
        code = orch.engine.wait(name, timeout_s=60)
        logs = orch.engine.logs(name).decode()
        assert code == 0, logs
        res = json.loads(logs.split("MITM ", 1)[1].splitlines()[0])
        assert res["ok"][0] == 200 and "/fine" in res["ok"][1], res
        assert res["denied"][0] == 403, res
    finally:
        orch.teardown(name, force=True)
        from clawker_amd.controlplane.client import CPClient
        CPClient(auto_start=False).stop()
        orch.close()


def test_mitm_stream_survives_slow_response(isolated_env, tmp_path, monkeypatch):
    """A response that stalls mid-body beyond the upstream CONNECT
    timeout (10s) must still complete through the MITM chain (SSE / slow
    LLM stream semantics; the relay read timeout is 300s)."""
    import http.server
    monkeypatch.setenv("CLAWKER_MITM_INSECURE_UPSTREAM", "1")
    key = tmp_path / "s.key"; crt = tmp_path / "s.crt"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "ec", "-pkeyopt",
         "ec_paramgen_curve:P-256", "-keyout", str(key), "-out", str(crt),
         "-nodes", "-subj", "/CN=slow.test", "-days", "2"],
        check=True, capture_output=True)

    class H(http.server.BaseHTTPRequestHandler):
        def do_GET(self):
            self.send_response(200)
            self.send_header("Content-Length", "9")
            self.end_headers()
            self.wfile.write(b"SLOW")
            self.wfile.flush()
            time.sleep(10.6)          # > connect timeout, < relay timeout
            self.wfile.write(b"_DONE")
        def log_message(self, *a):
            pass

    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), H)
    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
    ctx.load_cert_chain(str(crt), str(key))
    srv.socket = ctx.wrap_socket(srv.socket, server_side=True)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    port = srv.server_address[1]
    monkeypatch.setenv("CLAWKER_DNS_STATIC", "slow.test=127.0.0.1")
    from clawker_amd.config.schema import EgressRule
    from clawker_amd.firewall import EgressRulesStore, mitm
    from clawker_amd.firewall.gateway import GatewayManager
    from clawker_amd.firewall.policy import compile_policy, write_policy_snapshot
    EgressRulesStore().add([EgressRule(dst="slow.test", proto="tls",
                                       port=port, paths=["/"])])
    rundir = tmp_path / "rundir"
    rundir.mkdir()
    write_policy_snapshot(rundir, compile_policy())
    mgr = GatewayManager()
    mgr.attach("clawker.t.slow", rundir)
    try:
        raw = socket.socket(socket.AF_UNIX)
        raw.settimeout(30)
        raw.connect(str(rundir / "egress.sock"))
        raw.sendall(f"CONNECT slow.test:{port} HTTP/1.1\r\n\r\n".encode())
        assert b"200 Connection established" in raw.recv(100)
        ca_crt, _ = mitm.ensure_ca()
        cctx = ssl.create_default_context(cafile=str(ca_crt))
        tls = cctx.wrap_socket(raw, server_hostname="slow.test")
        tls.sendall(b"GET /stream HTTP/1.1\r\nHost: slow.test\r\n\r\n")
        data = b""
        t0 = time.monotonic()
        while b"SLOW_DONE" not in data:
            chunk = tls.recv(65536)
            assert chunk, f"stream died after {time.monotonic()-t0:.1f}s: {data!r}"
            data += chunk
        assert time.monotonic() - t0 >= 10.0     # it really stalled
        tls.close()
    finally:
        mgr.detach_all()
        srv.shutdown()
