"""HBM budget enforcement: env hints are not
enforcement — the CP watchdog samples drm fdinfo and kills on breach.
CPU tests cover the parsing + decision logic with injected samplers; the
gpu-marked test breaches a real budget with torch."""
import os
import time
from types import SimpleNamespace

import pytest


class FakeEvents:
    def __init__(self):
        self.records = []

    def emit(self, event, **kv):
        self.records.append({"event": event, **kv})


class FakeEngine:
    def __init__(self):
        self.killed = []

    def kill(self, name):
        self.killed.append(name)


def _info(name="sb", hbm_gb="1", gpus=(0,), pid=4242):
    return SimpleNamespace(name=name, pid=pid, gpus=list(gpus),
                           labels={"dev.clawker.hbm_gb": hbm_gb} if hbm_gb else {})


def test_watchdog_kill_on_breach():
    from clawker_amd.monitor.hbm import HBMWatchdog
    ev, eng = FakeEvents(), FakeEngine()
    usage = {"v": 2 << 30}    # 2 GiB used vs 1 GiB budget
    wd = HBMWatchdog(eng, ev, mode="kill", vram_fn=lambda i: usage["v"])
    wd.check([_info()])
    assert eng.killed == ["sb"]
    assert any(r["event"] == "hbm_budget_exceeded" and r["action"] == "kill"
               for r in ev.records)


def test_watchdog_warn_mode_and_hysteresis():
    from clawker_amd.monitor.hbm import HBMWatchdog
    ev, eng = FakeEvents(), FakeEngine()
    usage = {"v": int(0.95 * (1 << 30))}
    wd = HBMWatchdog(eng, ev, mode="warn", vram_fn=lambda i: usage["v"])
    wd.check([_info()])
    wd.check([_info()])     # second pass: no duplicate warning
    warns = [r for r in ev.records if r["event"] == "hbm_budget_warning"]
    assert len(warns) == 1
    assert not eng.killed
    # drop below 80% -> re-arm; cross again -> second warning
    usage["v"] = int(0.5 * (1 << 30))
    wd.check([_info()])
    usage["v"] = int(0.95 * (1 << 30))
    wd.check([_info()])
    warns = [r for r in ev.records if r["event"] == "hbm_budget_warning"]
    assert len(warns) == 2


def test_watchdog_budget_scales_with_gpu_count():
    from clawker_amd.monitor.hbm import HBMWatchdog
    ev, eng = FakeEvents(), FakeEngine()
    # 1.5 GiB used, 1 GiB/GPU budget, 2 GPUs -> within budget
    wd = HBMWatchdog(eng, ev, mode="kill",
                     vram_fn=lambda i: int(1.5 * (1 << 30)))
    wd.check([_info(gpus=(0, 1))])
    assert not eng.killed


def test_fdinfo_vram_parse_and_dedupe(tmp_path, monkeypatch):
    from clawker_amd.monitor import hbm
    # fake /proc: two pids sharing one drm client + one distinct
    proc = tmp_path / "proc"
    for pid, fds in {
        100: {"5": ("77", 1024)},          # client 77: 1 MiB
        101: {"7": ("77", 1024),           # same client inherited by fork
              "9": ("88", 2048)},          # client 88: 2 MiB
    }.items():
        d = proc / str(pid) / "fdinfo"
        d.mkdir(parents=True)
        for fd, (client, kib) in fds.items():
            (d / fd).write_text(
                f"pos: 0\ndrm-driver: amdgpu\ndrm-client-id: {client}\n"
                f"drm-memory-vram: {kib} KiB\n")
    monkeypatch.setattr(hbm, "PROC", proc)
    total = hbm.vram_bytes_for_pids([100, 101, 999])
    assert total == (1024 + 2048) * 1024   # client 77 counted once


def test_sandbox_pids_subtree():
    from clawker_amd.monitor.hbm import sandbox_pids
    me = os.getpid()
    import subprocess
    p = subprocess.Popen(["sleep", "5"])
    try:
        pids = sandbox_pids(me)
        assert me in pids
        assert p.pid in pids
    finally:
        p.kill()
        p.wait()


@pytest.mark.gpu
def test_hbm_breach_kills_sandbox_gpu(isolated_env, tmp_path):
    """e2e on hardware: a torch agent allocating past its budget is
    killed by the CP watchdog and the event is logged."""
    import yaml
    from clawker_amd import consts
    cfg_dir = consts.config_dir()
    cfg_dir.mkdir(parents=True, exist_ok=True)
    (cfg_dir / "settings.yaml").write_text(yaml.safe_dump(
        {"control_plane": {"drain_to_zero": False}}))
    ws = tmp_path / "hproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: hbmtest\n")
    from clawker_amd.config import load_config
    from clawker_amd.controlplane.client import CPClient
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(ws))
    # budget 2 GB; the agent tries to hold 8 GB and then idle
    payload = (
        "import torch, time; "
        "xs = [torch.empty(1024, 1024, 1024, device='cuda') for _ in range(8)]; "
        "torch.cuda.synchronize(); print('ALLOCATED', flush=True); "
        "time.sleep(120); print('SURVIVED', flush=True)")
    name = "clawker.hbmtest.breach"
    try:
        cp = CPClient()
        cp.ensure_running()
        orch.run(RunOptions(agent="breach", name=name, gpus=1, hbm_gb=2,
                            autostart=True, cmd=["python3", "-c", payload]))
        deadline = time.monotonic() + 90
        state = ""
        while time.monotonic() < deadline:
            state = orch.engine.inspect(name).state
            if state in ("exited", "dead"):
                break
            time.sleep(1)
        logs = orch.engine.logs(name).decode()
        assert state in ("exited", "dead"), \
            f"watchdog did not kill (state={state}): {logs[-400:]}"
        assert "SURVIVED" not in logs
        evs = cp.events(300)
        assert any(e.get("event") == "hbm_budget_exceeded"
                   and e.get("sandbox") == name for e in evs)
    finally:
        try:
            orch.teardown(name, force=True)
        except Exception:
            pass
        CPClient(auto_start=False).stop()
        orch.close()
