"""The flagship configuration, proven end to end: a `user: agent`
(non-root) sandbox whose agent can use the firewall gateway, its
bootstrap material, the MITM trust bundle and the host services — while
the admin surface (ctl.sock, spec.json, policy.json, rundir listing)
stays root-only.

Reference contract: clawkerd's STRICT listener + unprivileged spawn
(/root/reference/clawkerd/listener.go:145, spawn_unix.go:118); the
harness floors both declare `user: agent` (bundle assets), which the
engine materializes into the overlay upper (engine/users.py) since
hostfs sandboxes see the host's passwd.
"""
import json
import os
import threading
import time
from pathlib import Path

import pytest

from conftest import requires_isolation

pytestmark = requires_isolation


@pytest.fixture
def ws_orch(isolated_env, tmp_path):
    ws = tmp_path / "uproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: utest\n")
    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator
    orch = Orchestrator(load_config(ws))
    yield orch, ws
    for info in orch.engine.list():
        try:
            orch.teardown(info.name, force=True)
        except Exception:
            pass
    from clawker_amd.controlplane.client import CPClient
    CPClient(auto_start=False).stop()
    orch.close()


def _run_and_wait(orch, opts, timeout=60):
    from clawker_amd.orchestrator import RunOptions  # noqa: F401
    info = orch.run(opts)
    code = orch.engine.wait(info.name, timeout_s=timeout)
    logs = orch.engine.logs(info.name).decode()
    return info, code, logs


def test_named_user_materialized(ws_orch):
    """`user: agent` (the harness default) resolves in-sandbox to a
    non-root uid with a writable HOME and GPU group membership."""
    orch, ws = ws_orch
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.utest.id"
    script = ("echo UID=$(id -u) GID=$(id -g) USER=$(id -un); "
              "echo GROUPS=$(id -nG); echo HOME=$HOME; "
              "touch $HOME/probe && echo HOME_WRITABLE=yes")
    info, code, logs = _run_and_wait(orch, RunOptions(
        agent="id", name=name, autostart=True,
        cmd=["/bin/sh", "-c", script]))
    assert code == 0, logs
    assert "USER=agent" in logs
    assert "UID=0" not in logs.replace("UID=0 ", "UID=ZERO ")  # uid != 0
    uid = int(logs.split("UID=")[1].split()[0])
    assert uid >= 1000
    assert "HOME=/home/agent" in logs
    assert "HOME_WRITABLE=yes" in logs
    assert info.labels.get("dev.clawker.uid") == str(uid)
    # GPU device groups joined when the host defines them
    host_groups = Path("/etc/group").read_text()
    for g in ("render", "video"):
        if f"\n{g}:" in host_groups or host_groups.startswith(f"{g}:"):
            assert g in logs.split("GROUPS=")[1].splitlines()[0]


def test_workspace_owner_uid_adopted(ws_orch):
    """The materialized user takes the workspace owner's uid/gid so
    bind-mounted files stay writable (reference: host-UID user setup in
    Dockerfile.base.tmpl — without idmap mounts)."""
    orch, ws = ws_orch
    os.chown(ws, 4321, 4321)
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.utest.wsw"
    info, code, logs = _run_and_wait(orch, RunOptions(
        agent="wsw", name=name, autostart=True,
        cmd=["/bin/sh", "-c",
             "id -u; echo data > /workspace/agent-made.txt && echo WROTE=ok"]))
    assert code == 0, logs
    assert "4321" in logs
    assert "WROTE=ok" in logs
    made = ws / "agent-made.txt"
    assert made.exists()
    assert made.stat().st_uid == 4321


def test_admin_surface_root_only(ws_orch):
    """The rundir is traversable (0711) but the agent cannot enumerate
    it, read the spec/policy, or connect to the control socket."""
    orch, ws = ws_orch
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.utest.adm"
    script = r"""
import json, os, socket
out = {}
def probe(fn):
    try:
        return fn()
    except OSError as e:
        return f"denied:{e.errno}"
out["list_rundir"] = probe(lambda: os.listdir("/run/clawker"))
out["read_spec"] = probe(lambda: open("/run/clawker/spec.json").read()[:10])
out["read_token"] = probe(
    lambda: open("/run/clawker/bootstrap/token").read().strip()[:9])
def ctl():
    s = socket.socket(socket.AF_UNIX)
    s.connect("/run/clawker/ctl.sock")
    return "connected"
out["ctl_sock"] = probe(ctl)
print("RESULT " + json.dumps(out), flush=True)
"""
    info, code, logs = _run_and_wait(orch, RunOptions(
        agent="adm", name=name, autostart=True,
        cmd=["python3", "-c", script]))
    assert code == 0, logs
    res = json.loads(logs.split("RESULT ", 1)[1].splitlines()[0])
    assert str(res["list_rundir"]).startswith("denied:13")      # EACCES
    assert str(res["read_spec"]).startswith("denied:13")
    assert str(res["ctl_sock"]).startswith("denied:13")
    # the agent's own identity material IS readable
    assert res["read_token"].startswith(name.split(".")[0])  # "clawker..."


def test_exec_as_unknown_user_fails_loudly(ws_orch):
    """No silent root fallback: exec with an unresolvable named user is
    refused (reference: spawn_unix.go fails the spawn)."""
    orch, ws = ws_orch
    from clawker_amd.errors import ClawkerError
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.utest.noresolve"
    orch.run(RunOptions(agent="noresolve", name=name, autostart=False,
                        cmd=["sleep", "30"]))
    with orch.client(name) as c:
        with pytest.raises(ClawkerError, match="user not found"):
            c.exec([{"argv": ["id", "-u"], "user": "no-such-user-xyz"}])
        # control connection stays healthy; a good exec still works
        code, out, _ = c.exec([{"argv": ["id", "-u"], "user": "agent"}])
        assert code == 0
        assert int(out.decode().strip()) >= 1000
    orch.teardown(name, force=True)


class _Upstream:
    def __init__(self):
        import http.server

        class H(http.server.BaseHTTPRequestHandler):
            def do_GET(self):
                body = b"UPSTREAM_OK"
                self.send_response(200)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def log_message(self, *a):
                pass

        self.srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), H)
        self.port = self.srv.server_address[1]
        threading.Thread(target=self.srv.serve_forever, daemon=True).start()

    def stop(self):
        self.srv.shutdown()


def test_firewall_and_services_as_agent(ws_orch, monkeypatch):
    """The full non-root journey: DNS + gateway egress + MITM trust
    bundle + hostproxy + ssh-agent bridge, all from the agent uid."""
    up = _Upstream()
    monkeypatch.setenv("CLAWKER_DNS_STATIC", "allowed.test=127.0.0.1")
    orch, ws = ws_orch
    from clawker_amd.config.schema import EgressRule
    from clawker_amd.firewall import EgressRulesStore
    EgressRulesStore().add(
        [EgressRule(dst="allowed.test", proto="http", port=up.port)])

    # a host ssh-agent socket for the bridge
    import socket as _socket
    agent_dir = ws.parent / "sshagent"
    agent_dir.mkdir()
    host_agent = agent_dir / "agent.sock"
    lst = _socket.socket(_socket.AF_UNIX)
    lst.bind(str(host_agent))
    lst.listen(4)

    def echo_agent():
        while True:
            try:
                conn, _ = lst.accept()
            except OSError:
                return
            data = conn.recv(64)
            conn.sendall(b"SSH-AGENT:" + data)
            conn.close()

    threading.Thread(target=echo_agent, daemon=True).start()
    monkeypatch.setenv("SSH_AUTH_SOCK", str(host_agent))

    script = r"""
import json, os, socket, urllib.request
out = {}
out["uid"] = os.getuid()
try:
    with urllib.request.urlopen("http://allowed.test:%PORT%/ok", timeout=10) as r:
        out["egress"] = (r.status, r.read().decode())
except Exception as e:
    out["egress"] = (-1, str(e))
out["trust_bundle"] = open(os.environ["SSL_CERT_FILE"]).read()[:27]
s = socket.socket(socket.AF_UNIX)
try:
    s.connect(os.environ["SSH_AUTH_SOCK"])
    s.sendall(b"ping")
    out["ssh_agent"] = s.recv(64).decode()
except OSError as e:
    out["ssh_agent"] = f"fail:{e}"
h = socket.socket(socket.AF_UNIX)
try:
    h.connect("/run/clawker/hostproxy.sock")
    out["hostproxy"] = "connect_ok"
except OSError as e:
    out["hostproxy"] = f"fail:{e.errno}"
print("RESULT " + json.dumps(out), flush=True)
"""
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.utest.svc"
    info = orch.run(RunOptions(
        agent="svc", name=name, autostart=False, firewall=True,
        host_services=True,
        cmd=["python3", "-c", script.replace("%PORT%", str(up.port))]))
    # wait for gateway attach, then release the CMD
    rundir = info.rundir
    end = time.monotonic() + 10
    while time.monotonic() < end:
        if (rundir / "egress.sock").exists() and (rundir / "dns.sock").exists():
            break
        time.sleep(0.05)
    with orch.client(name) as c:
        c.agent_ready()
    code = orch.engine.wait(name, timeout_s=60)
    logs = orch.engine.logs(name).decode()
    up.stop()
    lst.close()
    assert code == 0, logs
    res = json.loads(logs.split("RESULT ", 1)[1].splitlines()[0])
    assert res["uid"] >= 1000
    assert res["egress"][0] == 200 and "UPSTREAM_OK" in res["egress"][1]
    assert res["trust_bundle"].startswith("-----BEGIN CERTIFICATE----")
    assert res["ssh_agent"] == "SSH-AGENT:ping"
    assert res["hostproxy"] == "connect_ok"
    orch.teardown(name, force=True)


def test_volume_chown_never_follows_symlinks(ws_orch, tmp_path):
    """Agent-controlled snapshot content may contain symlinks to host
    paths; the post-create ownership pass must lchown, never retarget
    through the link (found by self-review, r02)."""
    import os as _os
    orch, ws = ws_orch
    victim = tmp_path / "victim.txt"
    victim.write_text("host file")
    _os.chown(victim, 0, 0)
    _os.symlink(victim, ws / "link-to-host")
    (ws / "normal.txt").write_text("x")
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.utest.lchown"
    orch.run(RunOptions(agent="lchown", name=name, autostart=True,
                        workspace_mode="snapshot",
                        cmd=["/bin/sh", "-c", "ls /workspace >/dev/null"]))
    orch.engine.wait(name, timeout_s=30)
    st = _os.stat(victim)
    assert (st.st_uid, st.st_gid) == (0, 0), \
        "chown followed a symlink out of the snapshot volume!"
    orch.teardown(name, force=True)


def test_exec_defaults_to_sandbox_user(ws_orch):
    """`clawker exec` without -u runs as the sandbox's configured user
    (docker semantics), never silently root."""
    import subprocess
    import sys as _sys
    orch, ws = ws_orch
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.utest.execd"
    orch.run(RunOptions(agent="execd", name=name, autostart=False,
                        cmd=["sleep", "60"]))
    with orch.client(name) as c:
        c.agent_ready()
    import os as _os
    env = dict(_os.environ, PYTHONPATH=str(Path(__file__).parents[1]))
    r = subprocess.run(
        [_sys.executable, "-m", "clawker_amd", "exec", name, "--",
         "id", "-u"], capture_output=True, text=True, env=env, timeout=60)
    assert r.returncode == 0, r.stderr
    assert int(r.stdout.strip()) >= 1000, r.stdout
    r2 = subprocess.run(
        [_sys.executable, "-m", "clawker_amd", "exec", "-u", "root", name,
         "--", "id", "-u"], capture_output=True, text=True, env=env,
        timeout=60)
    assert r2.returncode == 0, r2.stderr
    assert r2.stdout.strip() == "0"
    orch.teardown(name, force=True)
