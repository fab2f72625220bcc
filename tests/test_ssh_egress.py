"""SSH / raw-TCP egress for non-proxy-aware clients.

The staged clawker-ssh-proxy ProxyCommand tunnels stdio through the
gateway's CONNECT endpoint under `proto: ssh` rules. The test talks a
fake ssh banner exchange through it (sshd isn't installed in CI; the
gateway is transport-level, so the byte relay IS the contract).

Reference: Envoy sequential TCP/SSH listeners (envoy_config.go) and the
VCS ssh egress merge (/root/reference/internal/cmd/project/init/
init.go:108-183).
"""
import json
import socket
import threading
import time

import pytest

from conftest import requires_isolation

pytestmark = requires_isolation


@pytest.fixture
def ssh_server():
    """Fake sshd: sends its banner, echoes one line back prefixed."""
    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(4)
    port = srv.getsockname()[1]

    def run():
        while True:
            try:
                c, _ = srv.accept()
            except OSError:
                return
            try:
                c.sendall(b"SSH-2.0-FakeSSHD\r\n")
                data = c.recv(256)
                c.sendall(b"GOT:" + data)
            except OSError:
                pass
            c.close()

    threading.Thread(target=run, daemon=True).start()
    yield port
    srv.close()


@pytest.fixture
def fw_orch(isolated_env, tmp_path, monkeypatch, ssh_server):
    monkeypatch.setenv("CLAWKER_DNS_STATIC", "sshhost.test=127.0.0.1")
    ws = tmp_path / "sproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: sshtest\n")
    from clawker_amd.config import load_config
    from clawker_amd.config.schema import EgressRule
    from clawker_amd.firewall import EgressRulesStore
    from clawker_amd.orchestrator import Orchestrator
    EgressRulesStore().add([
        EgressRule(dst="sshhost.test", proto="ssh", port=ssh_server)])
    orch = Orchestrator(load_config(ws))
    yield orch, ws, ssh_server
    for info in orch.engine.list():
        try:
            orch.teardown(info.name, force=True)
        except Exception:
            pass
    from clawker_amd.controlplane.client import CPClient
    CPClient(auto_start=False).stop()
    orch.close()


AGENT = r"""
set -e
printf 'probe-payload' | /run/clawker/bin/clawker-ssh-proxy sshhost.test %PORT% \
  > /tmp/ssh_ok.out 2>/tmp/ssh_ok.err; echo "ALLOWED_RC=$?"
cat /tmp/ssh_ok.out
/run/clawker/bin/clawker-ssh-proxy denied.test %PORT% </dev/null \
  > /tmp/ssh_no.out 2>/tmp/ssh_no.err || echo "DENIED_RC=$?"
cat /tmp/ssh_no.err
echo "GIT_SSH=$GIT_SSH_COMMAND"
"""


def _wait_gateway(orch, name, deadline=10.0):
    rundir = orch.engine.inspect(name).rundir
    end = time.monotonic() + deadline
    while time.monotonic() < end:
        if (rundir / "egress.sock").exists():
            return True
        time.sleep(0.05)
    return False


def test_ssh_proxycommand_tunnel(fw_orch):
    orch, ws, port = fw_orch
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.sshtest.agent"
    orch.run(RunOptions(
        agent="agent", name=name, autostart=False, firewall=True,
        cmd=["/bin/sh", "-c", AGENT.replace("%PORT%", str(port))]))
    assert _wait_gateway(orch, name)
    with orch.client(name) as c:
        c.agent_ready()
    code = orch.engine.wait(name, timeout_s=60)
    logs = orch.engine.logs(name).decode()
    assert code == 0, logs
    # allowed: full duplex through the tunnel (banner + echo)
    assert "ALLOWED_RC=0" in logs
    assert "SSH-2.0-FakeSSHD" in logs
    assert "GOT:probe-payload" in logs
    # denied: helper exits non-zero with a clear message
    assert "DENIED_RC=1" in logs
    assert "egress denied" in logs
    # git-over-ssh is wired to the helper
    assert "GIT_SSH=ssh -o ProxyCommand=" in logs
    assert "clawker-ssh-proxy" in logs.split("GIT_SSH=")[1]
    orch.teardown(name, force=True)


def test_init_git_protocol_ssh_merges_rules(isolated_env, tmp_path, monkeypatch):
    """`clawker init --vcs github --git-protocol ssh` writes proto-ssh
    egress rules next to the https ones."""
    import yaml
    from click.testing import CliRunner
    from clawker_amd.cli.root import cli
    proj = tmp_path / "initproj"
    proj.mkdir()
    monkeypatch.chdir(proj)
    r = CliRunner().invoke(cli, ["init", "--name", "sshinit", "--vcs",
                                 "github", "--git-protocol", "ssh", "-y"])
    assert r.exit_code == 0, r.output
    doc = yaml.safe_load((proj / ".clawker.yaml").read_text())
    rules = doc["security"]["egress"]
    assert {"dst": "github.com", "proto": "ssh", "port": 22} in rules
    assert any(r["proto"] == "tls" and r["dst"] == "github.com" for r in rules)
