"""CP-driven plans for detached sandboxes: a gated
sandbox whose starting client died still reaches AgentReady via the CP
watcher (reference: CP-side Executor, init_steps.go:67)."""
import time

import pytest

from conftest import requires_isolation

pytestmark = requires_isolation


def test_orphaned_gated_sandbox_boots_via_cp(isolated_env, tmp_path):
    import yaml
    from clawker_amd import consts
    # shrink the orphan grace so the test is fast
    cfg_dir = consts.config_dir()
    cfg_dir.mkdir(parents=True, exist_ok=True)
    (cfg_dir / "settings.yaml").write_text(yaml.safe_dump(
        {"control_plane": {"orphan_grace_s": 1, "drain_to_zero": False}}))

    ws = tmp_path / "oproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: otest\n")
    from clawker_amd.config import load_config
    from clawker_amd.controlplane.client import CPClient
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(ws))
    name = "clawker.otest.orphan"
    try:
        # gated start; the "client" (this test) deliberately never drives
        # plans or agent_ready and holds no ckd session
        orch.run(RunOptions(
            agent="orphan", name=name, autostart=False, firewall=False,
            cmd=["/bin/sh", "-c", "echo ORPHAN_BOOTED; exit 0"]))
        cp = CPClient()
        cp.ensure_running()
        # the CP watcher should drive Init/Boot and release the CMD
        code = orch.engine.wait(name, timeout_s=30)
        logs = orch.engine.logs(name).decode()
        assert code == 0, logs
        assert "ORPHAN_BOOTED" in logs
        evs = cp.events(200)
        assert any(e.get("event") == "cp_plans_driven"
                   and e.get("sandbox") == name for e in evs), \
            [e.get("event") for e in evs]
    finally:
        try:
            orch.teardown(name, force=True)
        except Exception:
            pass
        CPClient(auto_start=False).stop()
        orch.close()


def test_live_client_not_raced_by_cp(isolated_env, tmp_path):
    """A client holding its ckd session (clients>1) is left alone."""
    import yaml
    from clawker_amd import consts
    cfg_dir = consts.config_dir()
    cfg_dir.mkdir(parents=True, exist_ok=True)
    (cfg_dir / "settings.yaml").write_text(yaml.safe_dump(
        {"control_plane": {"orphan_grace_s": 1, "drain_to_zero": False}}))
    ws = tmp_path / "oproj2"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: otest2\n")
    from clawker_amd.config import load_config
    from clawker_amd.controlplane.client import CPClient
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(ws))
    name = "clawker.otest2.held"
    try:
        orch.run(RunOptions(
            agent="held", name=name, autostart=False, firewall=False,
            cmd=["/bin/sh", "-c", "echo HELD_BOOTED"]))
        cp = CPClient()
        cp.ensure_running()
        with orch.client(name) as c:        # live session held open
            time.sleep(4)                   # well past the orphan grace
            h = c.hello()
            assert not h.get("cmd_running"), \
                "CP drove plans despite a live client session"
            c.agent_ready()                 # the client releases it itself
        assert orch.engine.wait(name, timeout_s=30) == 0
    finally:
        try:
            orch.teardown(name, force=True)
        except Exception:
            pass
        CPClient(auto_start=False).stop()
        orch.close()
