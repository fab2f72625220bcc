"""Stats collection: per-sandbox cgroup metrics + per-GPU rocm telemetry
(replaces the reference's docker-stats stream, stats.go:194)."""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from pathlib import Path

from ..engine import Engine
from .sampler import GpuSample, RocmSampler

_CG_ROOT = Path("/sys/fs/cgroup")


def _read_int(p: Path) -> int | None:
    try:
        v = p.read_text().strip()
        return int(v) if v.isdigit() else None
    except OSError:
        return None


_cpu_prev: dict[str, tuple[float, int]] = {}   # name -> (ts, cpu_ns)


def cgroup_stats(name: str) -> dict:
    """Memory + pids + CPU%% for a sandbox's cgroup (v1 hybrid or v2)."""
    out: dict = {}
    cpu_ns = None
    v2 = _CG_ROOT / "clawker" / name
    if (v2 / "memory.current").exists():
        out["mem_bytes"] = _read_int(v2 / "memory.current")
        out["pids"] = _read_int(v2 / "pids.current")
        try:
            for line in (v2 / "cpu.stat").read_text().splitlines():
                if line.startswith("usage_usec"):
                    cpu_ns = int(line.split()[1]) * 1000
        except OSError:
            pass
    else:
        out["mem_bytes"] = _read_int(
            _CG_ROOT / "memory" / "clawker" / name / "memory.usage_in_bytes")
        out["pids"] = _read_int(
            _CG_ROOT / "pids" / "clawker" / name / "pids.current")
        cpu_ns = _read_int(_CG_ROOT / "cpuacct" / "clawker" / name / "cpuacct.usage")
    if cpu_ns is not None:
        now = time.monotonic()
        prev = _cpu_prev.get(name)
        _cpu_prev[name] = (now, cpu_ns)
        if prev and now > prev[0]:
            out["cpu_pct"] = round(
                100.0 * (cpu_ns - prev[1]) / ((now - prev[0]) * 1e9), 1)
    return out


@dataclass
class StatsSnapshot:
    ts: float = field(default_factory=time.time)
    sandboxes: list[dict] = field(default_factory=list)
    gpus: list[GpuSample] = field(default_factory=list)
    allocations: dict = field(default_factory=dict)   # gpu index -> sandbox
    events: list[dict] = field(default_factory=list)  # recent egress decisions


_sampler: RocmSampler | None = None


def _console_tail(rundir: Path, max_len: int = 72) -> str:
    try:
        with open(rundir / "console.log", "rb") as f:
            f.seek(0, 2)
            size = f.tell()
            f.seek(max(0, size - 4096))
            lines = [l for l in f.read().decode(errors="replace").splitlines()
                     if l.strip()]
            return lines[-1][-max_len:] if lines else ""
    except OSError:
        return ""


def collect_stats(engine: Engine, with_tails: bool = True) -> StatsSnapshot:
    global _sampler
    snap = StatsSnapshot()
    for info in engine.list():
        row = {
            "name": info.name, "state": info.state, "pid": info.pid,
            "gpus": info.gpus, "agent": info.agent, "project": info.project,
        }
        if info.state == "running":
            row.update(cgroup_stats(info.name))
            # per-sandbox VRAM attribution via drm fdinfo (the same
            # mechanism the HBM watchdog enforces with — monitor/hbm.py)
            if info.pid and info.gpus:
                try:
                    from .hbm import sandbox_pids, vram_bytes_for_pids
                    row["vram_bytes"] = vram_bytes_for_pids(
                        sandbox_pids(info.pid))
                except Exception:
                    pass
        if with_tails:
            row["tail"] = _console_tail(info.rundir)
        snap.sandboxes.append(row)
        for g in info.gpus:
            snap.allocations[g] = info.name
    try:
        if _sampler is None:
            _sampler = RocmSampler()
        snap.gpus = _sampler.sample()
    except Exception:
        snap.gpus = []
    try:
        from ..controlplane.daemon import EventLog, events_path
        snap.events = [e for e in EventLog(events_path()).tail(40)
                       if e.get("event") == "egress_decision"][-8:]
    except Exception:
        snap.events = []
    return snap


def render_stats(snap: StatsSnapshot):
    from rich.console import Group
    from rich.table import Table
    t = Table(title="sandboxes", box=None, pad_edge=False)
    for c in ("NAME", "STATE", "PID", "CPU", "MEM", "PIDS", "GPUS",
              "VRAM", "GPU-BUSY"):
        t.add_column(c)
    busy_by_idx = {g.index: g.busy_pct for g in snap.gpus}
    for s in snap.sandboxes:
        mem = s.get("mem_bytes")
        cpu = s.get("cpu_pct")
        vram = s.get("vram_bytes")
        busy = [busy_by_idx[i] for i in s["gpus"] if i in busy_by_idx]
        t.add_row(s["name"], s["state"], str(s.get("pid") or "-"),
                  f"{cpu:.0f}%" if cpu is not None else "-",
                  f"{mem / 2**20:.0f}M" if mem else "-",
                  str(s.get("pids") or "-"),
                  ",".join(map(str, s["gpus"])) or "-",
                  f"{vram / 2**30:.1f}G" if vram else "-",
                  f"{max(busy):.0f}%" if busy else "-")
    if not snap.gpus:
        return t
    g = Table(title="GPUs (MI355X)", box=None, pad_edge=False)
    for c in ("GPU", "BUSY", "VRAM", "POWER", "TEMP(J)", "SCLK", "OWNER"):
        g.add_column(c)
    for s in snap.gpus:
        g.add_row(str(s.index), f"{s.busy_pct:.0f}%",
                  f"{s.vram_used / 2**30:.1f}/{s.vram_total / 2**30:.0f}G",
                  f"{s.power_w:.0f}W", f"{s.temp_junction_c:.0f}C",
                  f"{s.sclk_mhz:.0f}MHz",
                  snap.allocations.get(s.index, "-"))
    return Group(t, g)
