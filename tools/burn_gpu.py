#!/usr/bin/env python3
"""GPU concurrency burn: W workers sustaining GPU-pinned agent loops
against the node's real GPU inventory for T seconds.

Each loop: allocate 1 GPU (retrying while the inventory is contended —
on a 1-GPU box 8 workers serialize on the allocator exactly like 8
agents queueing for a free device), cold-start a sandbox pinned to it,
run a torch matmul+backward on cuda:0 inside as an UNPRIVILEGED uid,
verify, tear down. Reports loop throughput, GPU-wait vs run split, and
leak counters — the "max concurrent GPU-pinned agent loops" half of
the BASELINE metric, measured not extrapolated.

Usage: python tools/burn_gpu.py [workers] [seconds] [out.json]
"""
from __future__ import annotations

import json
import os
import statistics
import sys
import threading
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def setup_dirs() -> Path:
    base = Path(os.environ.get("TMPDIR", "/tmp")) / "clawker-gpu-burn"
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


PAYLOAD = (
    "import os, torch; "
    "assert torch.cuda.is_available(); "
    "x = torch.randn(512, 512, device='cuda', requires_grad=True); "
    "(x @ x).sum().backward(); torch.cuda.synchronize(); "
    "print('BURN_OK uid=%d' % os.getuid(), flush=True)")


def main() -> int:
    workers = int(sys.argv[1]) if len(sys.argv) > 1 else 8
    seconds = float(sys.argv[2]) if len(sys.argv) > 2 else 300.0
    out_path = sys.argv[3] if len(sys.argv) > 3 else "/tmp/gpu-burn.json"
    setup_dirs()

    ws = Path(os.environ["CLAWKER_STATE_DIR"]) / "ws"
    ws.mkdir(exist_ok=True)
    (ws / ".clawker.yaml").write_text("project: burn\n")

    from clawker_amd.config import load_config
    from clawker_amd.gpu.allocator import GPUAllocationError
    from clawker_amd.orchestrator import Orchestrator, RunOptions

    orch = Orchestrator(load_config(ws))
    n_gpus = len(orch.allocator.inventory.devices)
    if n_gpus == 0:
        print(json.dumps({"error": "no GPUs on this host"}))
        return 2

    kfd_gid = os.stat("/dev/kfd").st_gid if os.path.exists("/dev/kfd") else 0
    user = f"54321:{kfd_gid}" if orch.engine.backend == "proc" else "agent"

    lock = threading.Lock()
    loops: list[dict] = []
    failures: list[str] = []
    stop_at = time.monotonic() + seconds

    def worker(w: int) -> None:
        i = 0
        while time.monotonic() < stop_at:
            i += 1
            name = f"clawker.burn.w{w}i{i}"
            t0 = time.perf_counter()
            # GPU wait: retry allocation while the inventory is contended
            while True:
                try:
                    info = orch.run(RunOptions(
                        agent=f"w{w}i{i}", name=name, gpus=1, autostart=True,
                        user=user, env={"HOME": "/tmp"}, hbm_gb=32,
                        cmd=["python3", "-c", PAYLOAD]))
                    break
                except GPUAllocationError:
                    if time.monotonic() >= stop_at:
                        return
                    time.sleep(0.05)
                except Exception as e:  # noqa: BLE001
                    with lock:
                        failures.append(f"{name} create: {e}")
                    return
            t1 = time.perf_counter()
            try:
                code = orch.engine.wait(name, timeout_s=180)
                logs = orch.engine.logs(name)
                ok = code == 0 and b"BURN_OK" in logs
                if not ok:
                    with lock:
                        failures.append(
                            f"{name}: exit={code} {logs[-200:]!r}")
            finally:
                try:
                    orch.teardown(name, force=True)
                except Exception as e:  # noqa: BLE001
                    with lock:
                        failures.append(f"{name} teardown: {e}")
            t2 = time.perf_counter()
            with lock:
                loops.append({"wait_ms": (t1 - t0) * 1000,
                              "run_ms": (t2 - t1) * 1000})

    ts = [threading.Thread(target=worker, args=(w,)) for w in range(workers)]
    t_start = time.perf_counter()
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    wall = time.perf_counter() - t_start

    waits = [l["wait_ms"] for l in loops]
    runs = [l["run_ms"] for l in loops]
    leaks = {
        "sandbox_rows": len(orch.engine.list()),
        "gpu_allocs": len(orch.allocator.allocations()),
    }
    from clawker_amd.controlplane.client import CPClient
    CPClient(auto_start=False).stop()
    orch.close()
    result = {
        "workers": workers,
        "gpus_on_node": n_gpus,
        "duration_s": round(wall, 1),
        "loops_completed": len(loops),
        "loops_per_min": round(len(loops) / wall * 60, 1),
        "failures": failures[:10],
        "failure_count": len(failures),
        "gpu_wait_ms": {"p50": round(statistics.median(waits), 1) if waits else None,
                        "p95": round(sorted(waits)[max(0, int(len(waits) * .95) - 1)], 1)
                               if waits else None},
        "loop_run_ms": {"p50": round(statistics.median(runs), 1) if runs else None,
                        "p95": round(sorted(runs)[max(0, int(len(runs) * .95) - 1)], 1)
                               if runs else None},
        "user": user,
        "isolation": orch.engine.backend,
        "leaks": leaks,
    }
    Path(out_path).write_text(json.dumps(result, indent=1))
    print(json.dumps(result))
    return 0 if not failures and not any(leaks.values()) else 1


if __name__ == "__main__":
    sys.exit(main())
