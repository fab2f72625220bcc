"""GPU allocation ledger: 1:1 pinning of MI355X devices to sandboxes.

Durable in the engine's sqlite DB so allocations survive CLI restarts and
are reclaimed when their sandbox is gone (the drift-guard idea of the
reference's firewall — INV-B2-016: never enforce against a stale owner —
applied to GPU ownership). Free-GPU allocation is the scheduler for the
worktree fan-out path (SURVEY.md §5 long-context analog: N agent loops,
one pinned GPU each).
"""
from __future__ import annotations

import json
import time

from ..errors import ClawkerError
from ..logger import get as get_logger
from .inventory import GPUInventory

log = get_logger("gpu")


class GPUAllocationError(ClawkerError):
    pass


_SCHEMA = """
CREATE TABLE IF NOT EXISTS gpu_alloc (
  gpu_index INTEGER PRIMARY KEY,
  sandbox TEXT NOT NULL,
  exclusive INTEGER NOT NULL DEFAULT 1,
  at REAL NOT NULL
);
"""


class GPUAllocator:
    def __init__(self, db, inventory: GPUInventory | None = None,
                 reserve: list[int] | None = None):
        """db: engine StateDB (shares its sqlite connection AND its
        transaction lock — concurrent allocate/release from fleet
        threads must not interleave execute/commit pairs)."""
        self._db = db.db
        self._statedb = db
        self._lock = db._lock
        self.inventory = inventory or GPUInventory.detect()
        self.reserve = set(reserve or [])
        self._db.executescript(_SCHEMA)
        self._db.commit()

    # -- queries ---------------------------------------------------------------
    def allocations(self) -> dict[int, str]:
        cur = self._db.execute("SELECT gpu_index, sandbox FROM gpu_alloc")
        return {int(r[0]): r[1] for r in cur.fetchall()}

    def free_indices(self) -> list[int]:
        used = set(self.allocations())
        return [d.index for d in self.inventory.devices
                if d.index not in used and d.index not in self.reserve]

    # -- mutation ---------------------------------------------------------------
    def reclaim_stale(self, live_sandboxes: set[str]) -> list[int]:
        """Free allocations whose sandbox no longer exists."""
        freed = []
        with self._lock:
            for idx, owner in self.allocations().items():
                if owner not in live_sandboxes:
                    self._db.execute(
                        "DELETE FROM gpu_alloc WHERE gpu_index=?", (idx,))
                    freed.append(idx)
            if freed:
                self._db.commit()
        if freed:
            log.info("gpu_reclaimed", indices=freed)
        return freed

    def allocate(self, sandbox: str, count: int,
                 prefer_xgmi_adjacent: bool = True,
                 explicit: list[int] | None = None) -> list[int]:
        """Atomically allocate `count` free GPUs to `sandbox`."""
        if count <= 0:
            return []
        with self._lock, self._db:   # transaction
            free = self.free_indices()
            if explicit is not None:
                missing = [i for i in explicit if i not in free]
                if missing:
                    raise GPUAllocationError(
                        f"requested GPUs unavailable: {missing} (free: {free})")
                chosen = list(explicit)
            elif len(free) < count:
                raise GPUAllocationError(
                    f"need {count} GPUs, only {len(free)} free "
                    f"(allocations: {json.dumps(self.allocations())})")
            elif count > 1 and prefer_xgmi_adjacent:
                chosen = self.inventory.xgmi_adjacent_set(count, free) or free[:count]
            else:
                chosen = free[:count]
            now = time.time()
            for idx in chosen:
                self._db.execute(
                    "INSERT INTO gpu_alloc (gpu_index, sandbox, exclusive, at) "
                    "VALUES (?,?,1,?)", (idx, sandbox, now))
        log.info("gpu_allocated", sandbox=sandbox, indices=chosen)
        return chosen

    def release(self, sandbox: str) -> list[int]:
        with self._lock:
            cur = self._db.execute(
                "SELECT gpu_index FROM gpu_alloc WHERE sandbox=?", (sandbox,))
            freed = [int(r[0]) for r in cur.fetchall()]
            self._db.execute("DELETE FROM gpu_alloc WHERE sandbox=?", (sandbox,))
            self._db.commit()
        if freed:
            log.info("gpu_released", sandbox=sandbox, indices=freed)
        return freed
