"""Orchestrator tests on CPU (no GPU allocation; real sandboxes)."""
import json
import os
import subprocess
import sys
from pathlib import Path

import pytest

from conftest import requires_isolation

REPO = Path(__file__).resolve().parent.parent


@pytest.fixture
def proj(isolated_env, tmp_path):
    ws = tmp_path / "proj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text(
        "project: otest\nagent:\n  env:\n    FROM_PROJECT: yes1\n")
    (ws / "data.txt").write_text("payload")
    return ws


@pytest.fixture
def orch(proj):
    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator
    o = Orchestrator(load_config(proj))
    yield o
    for info in o.engine.list():
        try:
            o.teardown(info.name, force=True)
        except Exception:
            pass
    o.close()


@requires_isolation
def test_run_env_contract_and_workspace(orch, proj):
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.otest.a1"
    orch.run(RunOptions(
        agent="a1", name=name, autostart=True,
        cmd=["/bin/sh", "-c",
             "echo P=$CLAWKER_PROJECT A=$CLAWKER_AGENT F=$CLAWKER_FIREWALL "
             "E=$FROM_PROJECT; cat /workspace/data.txt; pwd"]))
    assert orch.engine.wait(name, timeout_s=30) == 0
    out = orch.engine.logs(name).decode()
    assert "P=otest A=a1 F=1 E=yes1" in out
    assert "payload" in out
    assert "/workspace" in out        # default workdir = workspace mount


@requires_isolation
def test_snapshot_workspace_is_disposable(orch, proj):
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.otest.snap"
    orch.run(RunOptions(
        agent="snap", name=name, autostart=True, workspace_mode="snapshot",
        cmd=["/bin/sh", "-c", "echo scribble >> /workspace/data.txt; cat /workspace/data.txt"]))
    assert orch.engine.wait(name, timeout_s=30) == 0
    assert b"scribble" in orch.engine.logs(name)
    # host workspace untouched (reference: SnapshotStrategy disposable copy)
    assert (proj / "data.txt").read_text() == "payload"
    orch.teardown(name, force=True)


@requires_isolation
def test_no_firewall_shares_host_netns(orch):
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.otest.hostnet"
    orch.run(RunOptions(agent="hostnet", name=name, autostart=True, firewall=False,
                        cmd=["/bin/sh", "-c", "ls /sys/class/net"]))
    assert orch.engine.wait(name, timeout_s=30) == 0
    out = orch.engine.logs(name).decode()
    assert "eth0" in out or "ens" in out or "enp" in out


def test_gpu_request_fails_without_gpus(orch):
    from clawker_amd.gpu import GPUAllocationError
    from clawker_amd.orchestrator import RunOptions
    if orch.allocator.inventory.devices:
        pytest.skip("host has GPUs")
    with pytest.raises(GPUAllocationError):
        orch.create(RunOptions(agent="g", name="clawker.otest.g", gpus=1, cmd=["true"]))


@requires_isolation
def test_bench_cpu_single():
    env = dict(os.environ)
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR", "MASTER_PORT"):
        env.pop(k, None)
    r = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=300, env=env, cwd=str(REPO))
    assert r.returncode == 0, r.stderr[-3000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["unit"] == "ms"
    assert out["steps"] == 2
    assert out["data"] == "synthetic"
    assert out["config"]["gpu_pinned"] is False      # CPU box


@requires_isolation
def test_bench_distributed_gloo_world2():
    """The driver launches bench via torch.distributed.run for N>1;
    exercise that path with gloo on CPU (world_size=2)."""
    env = dict(os.environ)
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR", "MASTER_PORT"):
        env.pop(k, None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", str(REPO / "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "0"],
        capture_output=True, text=True, timeout=600, env=env, cwd=str(REPO))
    assert r.returncode == 0, (r.stdout[-2000:], r.stderr[-3000:])
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 2
    assert out["config"]["concurrent_loops"] == 2


@requires_isolation
def test_clawkerignore_respected_in_snapshots(orch, proj):
    from clawker_amd.orchestrator import RunOptions
    (proj / ".clawkerignore").write_text("# comment\n.git/\n*.secret\nbuild/\n")
    (proj / ".git").mkdir()
    (proj / ".git" / "HEAD").write_text("ref: x")
    (proj / "api.secret").write_text("k")
    (proj / "build").mkdir()
    (proj / "build" / "out.o").write_text("o")
    (proj / "keep.txt").write_text("keep")
    name = "clawker.otest.ign"
    orch.run(RunOptions(agent="ign", name=name, autostart=True,
                        workspace_mode="snapshot", firewall=False,
                        cmd=["/bin/sh", "-c", "ls -a /workspace"]))
    assert orch.engine.wait(name, timeout_s=30) == 0
    out = orch.engine.logs(name).decode()
    assert "keep.txt" in out and "data.txt" in out
    assert ".git" not in out and "api.secret" not in out and "build" not in out
    orch.teardown(name, force=True)


def test_snapshot_excludes_nested_clawker_dirs(isolated_env, tmp_path, monkeypatch):
    """A data dir nested INSIDE the workspace must not be snapshot-copied
    (regression: self-copy recursion until ENAMETOOLONG)."""
    import os
    ws = tmp_path / "nested"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: nested\n")
    (ws / "code.txt").write_text("x")
    # relocate the clawker dirs INSIDE the workspace (plausible user setup)
    for var, sub in (("CLAWKER_DATA_DIR", "d"), ("CLAWKER_STATE_DIR", "s")):
        monkeypatch.setenv(var, str(ws / sub))
    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(ws))
    name = "clawker.nested.a"
    try:
        orch.run(RunOptions(agent="a", name=name, autostart=True,
                            firewall=False, host_services=False,
                            workspace_mode="snapshot",
                            cmd=["/bin/sh", "-c", "ls; test ! -e d && test ! -e s"]))
        assert orch.engine.wait(name, timeout_s=30) == 0, \
            orch.engine.logs(name).decode()
        vol = orch.engine.ensure_volume(f"{name}-snapshot", {})[0]
        assert (vol / "code.txt").is_file()
        assert not (vol / "d").exists() and not (vol / "s").exists()
    finally:
        orch.teardown(name, force=True)
        orch.close()


def test_worktree_snapshot_mode_refused(isolated_env, tmp_path):
    """Snapshot mode over a git worktree is refused (the copy's .git file
    would dangle and the agent's commits would be stranded)."""
    import subprocess
    repo = tmp_path / "wtrepo"
    repo.mkdir()
    (repo / ".clawker.yaml").write_text("project: wtsnap\n")
    (repo / "f").write_text("x")
    subprocess.run(["git", "init", "-q", "-b", "main"], cwd=repo, check=True)
    subprocess.run(["git", "add", "-A"], cwd=repo, check=True)
    subprocess.run(["git", "-c", "user.email=t@t", "-c", "user.name=t",
                    "commit", "-q", "-m", "i"], cwd=repo, check=True)
    wt = tmp_path / "wt"
    subprocess.run(["git", "worktree", "add", "-q", "-b", "b1", str(wt)],
                   cwd=repo, check=True)
    from clawker_amd.config import load_config
    from clawker_amd.errors import ClawkerError
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(repo))
    try:
        with pytest.raises(ClawkerError, match="worktree"):
            orch.create(RunOptions(agent="a", workspace=wt,
                                   workspace_mode="snapshot", firewall=False,
                                   cmd=["/bin/true"]))
        # and the failed create leaked nothing
        assert orch.engine.list() == []
    finally:
        orch.close()
