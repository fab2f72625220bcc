"""Fleet fan-out tests: N parallel agent loops over worktrees (CPU; the
GPU pinning branch is covered by allocator tests + gpu-marked tests)."""
import json
import subprocess
import time

import pytest

from conftest import requires_isolation

pytestmark = requires_isolation


@pytest.fixture
def git_proj(isolated_env, tmp_path):
    root = tmp_path / "fleetproj"
    root.mkdir()
    (root / ".clawker.yaml").write_text(
        "project: fleettest\nagent:\n  harness: echo\n")
    (root / "work.txt").write_text("base\n")
    subprocess.run(["git", "init", "-q", "-b", "main"], cwd=root, check=True)
    subprocess.run(["git", "add", "-A"], cwd=root, check=True)
    subprocess.run(["git", "-c", "user.email=t@t", "-c", "user.name=t",
                    "commit", "-q", "-m", "init"], cwd=root, check=True)
    return root


def test_fleet_up_wait_down_with_worktrees(git_proj):
    from clawker_amd.config import load_config
    from clawker_amd.fleet import Fleet, FleetOptions
    cfg = load_config(git_proj)
    fleet = Fleet(cfg)
    members = fleet.up(FleetOptions(
        count=3, firewall=False,
        cmd=["/bin/sh", "-c",
             "echo agent=$CLAWKER_AGENT branch=$(git branch --show-current); "
             "echo done-$CLAWKER_AGENT >> out.txt"]))
    assert len(members) == 3
    assert {m.branch for m in members} == {"agent/0", "agent/1", "agent/2"}
    fleet.wait(members, timeout_s=60)
    assert all(m.exit_code == 0 for m in members), [
        (m.sandbox, m.exit_code, fleet.orch.engine.logs(m.sandbox)[-200:])
        for m in members]
    # each agent worked in ITS OWN worktree
    for i, m in enumerate(members):
        logs = fleet.orch.engine.logs(m.sandbox).decode()
        assert f"branch=agent/{i}" in logs
    # the base repo is untouched
    assert (git_proj / "work.txt").read_text() == "base\n"
    assert not (git_proj / "out.txt").exists()
    # worktree dirs got the agent outputs
    from clawker_amd.project.worktrees import worktrees_dir
    for i in range(3):
        wt = worktrees_dir(cfg) / f"agent-{i}"
        assert (wt / "out.txt").read_text() == f"done-agent{i}\n"
    n = fleet.down()
    assert n == 3
    assert fleet.status() == []
    fleet.orch.close()


def test_fleet_snapshot_mode_without_git(isolated_env, tmp_path):
    root = tmp_path / "nogit"
    root.mkdir()
    (root / ".clawker.yaml").write_text("project: nogit\nagent:\n  harness: echo\n")
    (root / "seed.txt").write_text("s")
    from clawker_amd.config import load_config
    from clawker_amd.fleet import Fleet, FleetOptions
    fleet = Fleet(load_config(root))
    members = fleet.up(FleetOptions(
        count=2, firewall=False,
        cmd=["/bin/sh", "-c", "echo x >> seed.txt; cat seed.txt"]))
    fleet.wait(members, timeout_s=60)
    assert all(m.exit_code == 0 for m in members)
    assert (root / "seed.txt").read_text() == "s"   # snapshots, not binds
    fleet.down()
    fleet.orch.close()


def test_fleet_8way_worktrees_with_firewall(git_proj):
    """BASELINE config 3 shape: 8 concurrent worktree sandboxes with the
    egress firewall enabled (netns + gateways), one agent loop each."""
    from clawker_amd.config import load_config
    from clawker_amd.fleet import Fleet, FleetOptions
    cfg = load_config(git_proj)
    fleet = Fleet(cfg)
    members = fleet.up(FleetOptions(
        count=8, firewall=True, branch_prefix="w",
        cmd=["/bin/sh", "-c",
             "python3 -c \"import socket,sys\n"
             "s=socket.socket(); s.settimeout(2)\n"
             "try: s.connect(('203.0.113.9',443)); sys.exit(1)\n"
             "except OSError: sys.exit(0)\" "
             "&& echo FW_OK $CLAWKER_AGENT uid=$(id -u)"]))
    assert len(members) == 8
    fleet.wait(members, timeout_s=120)
    failures = [(m.sandbox, m.exit_code) for m in members if m.exit_code != 0]
    assert not failures, failures
    for i, m in enumerate(members):
        logs = fleet.orch.engine.logs(m.sandbox).decode()
        assert f"FW_OK w{i}" in logs
        # r02: fleet agents run as the materialized non-root user
        assert "uid=0" not in logs, logs
        # gateway sockets were attached for every member
        rundir = fleet.orch.engine.inspect(m.sandbox).rundir
        assert (rundir / "policy.json").exists()
    fleet.down(branch_prefix="w")
    from clawker_amd.controlplane.client import CPClient
    CPClient(auto_start=False).stop()
    fleet.orch.close()


def test_fleet_prompt_file(isolated_env, tmp_path):
    """fleet --prompt: each agent gets the prompt via the harness's
    prompt_cmd (the autonomous-loop driver shape, BASELINE config 4)."""
    root = tmp_path / "pfproj"
    root.mkdir()
    (root / ".clawker.yaml").write_text("project: pftest\nagent:\n  harness: echo\n")
    (root / "task.md").write_text("refactor the widget\n")
    from clawker_amd.config import load_config
    from clawker_amd.fleet import Fleet, FleetOptions
    fleet = Fleet(load_config(root))
    members = fleet.up(FleetOptions(
        count=2, firewall=False, use_worktrees=False,
        prompt_file=str(root / "task.md")))
    fleet.wait(members, timeout_s=60)
    for m in members:
        assert m.exit_code == 0
        logs = fleet.orch.engine.logs(m.sandbox).decode()
        assert "PROMPT:" in logs and "refactor the widget" in logs
    fleet.down()
    fleet.orch.close()


def test_fleet_run_one_shot(isolated_env, tmp_path, monkeypatch):
    """`clawker fleet run` = up + wait + report + down in one verb."""
    root = tmp_path / "oneshot"
    root.mkdir()
    (root / ".clawker.yaml").write_text("project: oneshot\nagent:\n  harness: echo\n")
    monkeypatch.chdir(root)
    from click.testing import CliRunner
    from clawker_amd.cli.root import cli
    r = CliRunner().invoke(cli, [
        "fleet", "run", "-n", "2", "--no-worktrees", "--no-firewall",
        "--timeout", "60", "--", "/bin/sh", "-c", "echo ran-$CLAWKER_AGENT"])
    if r.exception is not None and not isinstance(r.exception, SystemExit):
        raise r.exception
    assert r.exit_code == 0, r.output
    assert "exit=0" in r.output
    # sandboxes were torn down (no --keep)
    from clawker_amd.engine import Engine
    eng = Engine()
    assert [i for i in eng.list() if i.project == "oneshot"] == []
    eng.close()
    # failing member propagates worst exit code
    r = CliRunner().invoke(cli, [
        "fleet", "run", "-n", "1", "--no-worktrees", "--no-firewall",
        "--timeout", "60", "--", "/bin/sh", "-c", "exit 3"])
    from clawker_amd.errors import ClawkerError
    if isinstance(r.exception, ClawkerError):
        r.exit_code = r.exception.exit_code
    assert r.exit_code == 3, r.output
