"""Live ROCm telemetry via the native zero-spawn sampler.

Replaces the reference's `docker stats` streaming loop (stats.go:194
streamStats) and its compose monitoring stack's per-container metrics with
direct amdgpu sysfs sampling through clawker_amd._native (pre-opened fds,
microseconds per tick across 8 GPUs).

The native extension is REQUIRED on a GPU machine: if amdgpu devices are
present but the extension is missing, we raise instead of silently
degrading to a slow path.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field

from ..errors import ClawkerError
from ..gpu.inventory import GPUInventory


@dataclass
class GpuSample:
    index: int
    render_minor: int
    busy_pct: float = 0.0
    vram_used: int = 0
    vram_total: int = 0
    gtt_used: int = 0
    temp_edge_c: float = 0.0
    temp_junction_c: float = 0.0
    temp_mem_c: float = 0.0
    power_w: float = 0.0
    sclk_mhz: float = 0.0
    mclk_mhz: float = 0.0
    ts: float = field(default_factory=time.time)


class RocmSampler:
    def __init__(self, inventory: GPUInventory | None = None):
        self.inventory = inventory or GPUInventory.detect()
        self._minor_to_index = {d.render_minor: d.index for d in self.inventory.devices}
        try:
            from .. import _native
        except ImportError as e:
            if self.inventory.devices:
                raise ClawkerError(
                    "clawker_amd._native extension not built but amdgpu devices "
                    "present — run `make pymod` (no silent fallback on GPU nodes)"
                ) from e
            self._native = None
            return
        self._native = _native.GpuSampler(
            [d.render_minor for d in self.inventory.devices])

    def sample(self) -> list[GpuSample]:
        if self._native is None:
            return []
        out = []
        now = time.time()
        for row in self._native.sample():
            minor = int(row.get("minor", -1))
            s = GpuSample(index=self._minor_to_index.get(minor, -1),
                          render_minor=minor, ts=now)
            for k in ("busy_pct", "temp_edge_c", "temp_junction_c", "temp_mem_c",
                      "power_w", "sclk_mhz", "mclk_mhz"):
                if k in row:
                    setattr(s, k, float(row[k]))
            for k in ("vram_used", "vram_total", "gtt_used"):
                if k in row:
                    setattr(s, k, int(row[k]))
            out.append(s)
        return out
