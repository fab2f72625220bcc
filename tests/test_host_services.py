"""Hostproxy + socketbridge tests (reference: internal/hostproxy routes,
internal/socketbridge forwarding)."""
import json
import os
import socket
import threading
import time
from pathlib import Path

import pytest

from conftest import requires_isolation


@pytest.fixture
def hostproxy(isolated_env):
    from clawker_amd.hostproxy import HostProxyManager
    mgr = HostProxyManager()
    mgr.ensure_running()
    yield mgr
    mgr.stop()


def test_hostproxy_healthz_and_idempotent_ensure(hostproxy):
    status, body = hostproxy.request("GET", "/healthz")
    assert status == 200
    hostproxy.ensure_running()    # no second spawn
    assert hostproxy.running()


def test_open_url_policy_fail_closed(hostproxy, isolated_env):
    # no rule for the domain -> 403
    status, body = hostproxy.request(
        "POST", "/open/url", json.dumps({"url": "https://evil.example/x"}).encode())
    assert status == 403
    # rule added -> allowed (no browser on CI: opened:false with hint)
    from clawker_amd.config.schema import EgressRule
    from clawker_amd.firewall import EgressRulesStore
    EgressRulesStore().add([EgressRule(dst="docs.example", proto="tls", port=443,
                                       deny_paths=["/private"])])
    status, body = hostproxy.request(
        "POST", "/open/url", json.dumps({"url": "https://docs.example/page"}).encode())
    assert status == 200, body
    # deny_path enforced on open/url too (mirrored semantics)
    status, _ = hostproxy.request(
        "POST", "/open/url",
        json.dumps({"url": "https://docs.example/private/key"}).encode())
    assert status == 403
    # junk url
    status, _ = hostproxy.request(
        "POST", "/open/url", json.dumps({"url": "file:///etc/passwd"}).encode())
    assert status == 400


def test_callback_register_hit_and_poll(hostproxy):
    status, body = hostproxy.request(
        "POST", "/callback/register",
        json.dumps({"port": 8765, "sandbox": "clawker.t.a"}).encode())
    assert status == 200
    sess = json.loads(body)
    sid = sess["session"]
    assert f"/cb/{sid}/" in sess["callback_url"]
    # nothing captured yet
    status, body = hostproxy.request("GET", f"/callback/poll/{sid}")
    assert status == 200 and json.loads(body)["hits"] == []
    # the user's browser hits the callback
    status, body = hostproxy.request("GET", f"/cb/{sid}/done?code=xyz")
    assert status == 200 and b"authentication complete" in body
    # the agent polls and retrieves the redirect (incl. the auth code)
    status, body = hostproxy.request("GET", f"/callback/poll/{sid}")
    hits = json.loads(body)["hits"]
    assert len(hits) == 1 and hits[0]["path"] == "/done?code=xyz"
    status, _ = hostproxy.request("GET", "/cb/bogus/done")
    assert status == 404
    status, _ = hostproxy.request("GET", "/callback/poll/bogus")
    assert status == 404


def test_socketbridge_relays_to_host_agent(isolated_env, tmp_path, monkeypatch):
    """Fake host ssh-agent (unix echo) -> bridge in rundir -> client."""
    host_agent = tmp_path / "agent.sock"
    srv = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    srv.bind(str(host_agent))
    srv.listen(4)

    def serve():
        while True:
            try:
                c, _ = srv.accept()
            except OSError:
                return
            data = c.recv(1024)
            c.sendall(b"AGENT:" + data)
            c.close()

    threading.Thread(target=serve, daemon=True).start()
    monkeypatch.setenv("SSH_AUTH_SOCK", str(host_agent))

    from clawker_amd.socketbridge import SocketBridgeManager
    rundir = tmp_path / "rundir"
    rundir.mkdir()
    mgr = SocketBridgeManager()
    created = mgr.attach("clawker.t.sb", rundir)
    assert "ssh-agent.sock" in created
    c = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    c.settimeout(5)
    c.connect(str(rundir / "ssh-agent.sock"))
    c.sendall(b"sign-request")
    assert c.recv(100) == b"AGENT:sign-request"
    c.close()
    mgr.detach("clawker.t.sb")
    srv.close()


@requires_isolation
def test_hostproxy_reachable_from_inside_sandbox(isolated_env, tmp_path, hostproxy):
    """ns backend: the hostproxy socket is bind-mounted into the sandbox
    and the host-open helper talks through it (policy-denied URL -> 403)."""
    ws = tmp_path / "hsproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: hstest\n")
    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(ws))
    name = "clawker.hstest.a"
    try:
        orch.run(RunOptions(
            agent="a", name=name, autostart=True, firewall=True,
            host_services=True,
            cmd=["/bin/sh", "-c",
                 "ls -la /run/clawker/hostproxy.sock; "
                 "/run/clawker/bin/host-open.sh https://not-allowed.example/; "
                 "echo open_rc=$?"]))
        code = orch.engine.wait(name, timeout_s=30)
        logs = orch.engine.logs(name).decode()
        assert code == 0, logs
        assert "hostproxy.sock" in logs
        assert "open_rc=1" in logs       # denied by policy through the proxy
        assert "not in egress policy" in logs
    finally:
        orch.teardown(name, force=True)
        orch.close()


def test_hostproxy_crash_respawn(isolated_env):
    """A killed hostproxy is detected and respawned by ensure_running
    (failure-detection contract: subsystems recover, never wedge)."""
    import os
    import signal as _sig
    from clawker_amd.hostproxy import HostProxyManager
    mgr = HostProxyManager()
    mgr.ensure_running()
    assert mgr.running()
    # find and kill the daemon
    from clawker_amd import consts
    pid = int((consts.runtime_dir() / "hostproxy.pid").read_text())
    os.kill(pid, _sig.SIGKILL)
    # the killed daemon lingers as a zombie of this test process, so
    # liveness is judged by health, not kill(pid, 0)
    deadline = time.time() + 5
    while time.time() < deadline and mgr._healthy():
        time.sleep(0.02)
    assert not mgr._healthy()
    mgr.ensure_running()          # respawn, same socket path
    status, _ = mgr.request("GET", "/healthz")
    assert status == 200
    mgr.stop()
