#!/usr/bin/env python3
"""Soak test: sustained mixed workload against one node.

- N_LOOPS sequential cold-start loops (echo agents)
- 3 long-running background agents alive throughout
- one GPU-holding agent for the duration (when a GPU exists), then a
  final GPU cold start verifying allocator release
- leak tracking: sandbox rows, gpu allocations, runtime dirs, volumes,
  this process's fd count and children

Usage: python tools/soak.py [n_loops] [out.json]
"""
from __future__ import annotations

import json
import os
import statistics
import sys
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def setup_dirs() -> Path:
    base = Path(os.environ.get("TMPDIR", "/tmp")) / "clawker-soak"
    for var, sub in [
        ("CLAWKER_CONFIG_DIR", "config"), ("CLAWKER_DATA_DIR", "data"),
        ("CLAWKER_STATE_DIR", "state"), ("CLAWKER_RUNTIME_DIR", "run"),
        ("CLAWKER_IMAGE_DIR", "images"), ("CLAWKER_SANDBOX_DIR", "sandboxes"),
        ("CLAWKER_VOLUME_DIR", "volumes"),
    ]:
        d = base / sub
        d.mkdir(parents=True, exist_ok=True)
        os.environ[var] = str(d)
    return base


def fd_count() -> int:
    try:
        return len(os.listdir("/proc/self/fd"))
    except OSError:
        return -1


def main() -> int:
    n_loops = int(sys.argv[1]) if len(sys.argv) > 1 else 100
    out_path = Path(sys.argv[2]) if len(sys.argv) > 2 else Path("soak.json")
    base = setup_dirs()
    ws = base / "ws"
    ws.mkdir(exist_ok=True)
    (ws / ".clawker.yaml").write_text(
        "project: soak\nagent:\n  harness: echo\n"
        "workspace:\n  share_volume: false\n")

    import torch
    have_gpu = torch.cuda.is_available()

    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(ws))
    report: dict = {"n_loops": n_loops, "gpu": have_gpu,
                    "backend": orch.engine.backend, "errors": []}

    # long-running background agents
    bg = []
    for i in range(3):
        name = f"clawker.soak.bg{i}"
        orch.run(RunOptions(agent=f"bg{i}", name=name, autostart=True,
                            firewall=False, cmd=["sleep", "infinity"]))
        bg.append(name)
    gpu_holder = None
    if have_gpu:
        gpu_holder = "clawker.soak.gpuhold"
        orch.run(RunOptions(
            agent="gpuhold", name=gpu_holder, gpus=1, autostart=True,
            firewall=False,
            cmd=["python3", "-c",
                 "import torch, time; x=torch.randn(4096,4096,device='cuda',"
                 "dtype=torch.bfloat16); t0=time.time()\n"
                 "while time.time()-t0 < 3600: y = x @ x\n"]))

    fd0 = fd_count()
    lats = []
    t0 = time.time()
    for i in range(n_loops):
        name = f"clawker.soak.l{i}"
        s = time.perf_counter()
        try:
            orch.run(RunOptions(agent=f"l{i}", name=name, autostart=False,
                                firewall=False, gpus=0,
                                cmd=["/bin/sh", "-c", "echo done"]))
            with orch.client(name) as c:
                c.agent_ready()
            lats.append((time.perf_counter() - s) * 1000)
            code = orch.engine.wait(name, timeout_s=60)
            if code != 0:
                report["errors"].append(f"loop {i}: exit {code}")
        except Exception as e:
            report["errors"].append(f"loop {i}: {type(e).__name__}: {e}")
        finally:
            try:
                orch.teardown(name, force=True)
            except Exception as e:
                report["errors"].append(f"loop {i} teardown: {e}")
    elapsed = time.time() - t0

    # background agents still healthy?
    for name in bg:
        info = orch.engine.inspect(name)
        if info.state != "running":
            report["errors"].append(f"{name} died: {info.state}/{info.exit_code}")
        orch.teardown(name, force=True)
    if gpu_holder:
        info = orch.engine.inspect(gpu_holder)
        if info.state != "running":
            report["errors"].append(f"gpu holder died: {info.exit_code}")
        orch.teardown(gpu_holder, force=True)
        # allocator must be clean; a fresh GPU cold start must succeed
        if orch.allocator.allocations():
            report["errors"].append(f"gpu leak: {orch.allocator.allocations()}")
        name = "clawker.soak.gpufinal"
        orch.run(RunOptions(agent="gpufinal", name=name, gpus=1, autostart=True,
                            firewall=False, cmd=["/bin/sh", "-c", "ls /dev/kfd"]))
        if orch.engine.wait(name, timeout_s=60) != 0:
            report["errors"].append("final gpu cold start failed")
        orch.teardown(name, force=True)

    report.update({
        "elapsed_s": round(elapsed, 1),
        "loops_per_min": round(n_loops * 60 / elapsed, 1),
        "cold_start_ms": {
            "p50": round(statistics.median(lats), 2) if lats else None,
            "p95": round(sorted(lats)[int(len(lats) * 0.95) - 1], 2) if lats else None,
            "max": round(max(lats), 2) if lats else None,
        },
        "leaks": {
            "sandbox_rows": len(orch.engine.db.list_sandboxes()),
            "volumes": len(orch.engine.db.list_volumes()),
            "rundirs": len(list((base / "run" / "sandboxes").glob("*"))),
            "gpu_allocations": orch.allocator.allocations(),
            "fd_delta": fd_count() - fd0,
        },
    })
    orch.close()
    out_path.write_text(json.dumps(report, indent=1))
    print(json.dumps(report, indent=1))
    ok = (not report["errors"]
          and report["leaks"]["sandbox_rows"] == 0
          and report["leaks"]["rundirs"] == 0
          and abs(report["leaks"]["fd_delta"]) <= 8)
    print("SOAK", "PASS" if ok else "FAIL")
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
