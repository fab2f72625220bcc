"""GPU allocator unit tests with synthetic inventories (the real-GPU
paths live in test_gpu.py; these cover policy logic on CPU)."""
import pytest

from clawker_amd.engine.state import StateDB
from clawker_amd.gpu import GPUAllocationError, GPUAllocator
from clawker_amd.gpu.inventory import GPUDevice, GPUInventory


def _inv(n=8, xgmi_full=True):
    devs = [GPUDevice(index=i, render_minor=128 + i, card=i) for i in range(n)]
    if xgmi_full:
        for d in devs:
            d.xgmi_peers = [p.index for p in devs if p.index != d.index]
    return GPUInventory(devs)


@pytest.fixture
def alloc(isolated_env):
    db = StateDB()
    yield GPUAllocator(db, inventory=_inv())
    db.close()


def test_exclusive_allocation_and_release(alloc):
    a = alloc.allocate("clawker.t.a", 2)
    b = alloc.allocate("clawker.t.b", 2)
    assert len(a) == 2 and len(b) == 2 and not set(a) & set(b)
    assert len(alloc.free_indices()) == 4
    assert sorted(alloc.release("clawker.t.a")) == sorted(a)
    assert len(alloc.free_indices()) == 6


def test_exhaustion_raises(alloc):
    alloc.allocate("clawker.t.big", 8)
    with pytest.raises(GPUAllocationError):
        alloc.allocate("clawker.t.more", 1)


def test_explicit_indices_and_conflicts(alloc):
    got = alloc.allocate("clawker.t.x", 2, explicit=[3, 5])
    assert got == [3, 5]
    with pytest.raises(GPUAllocationError):
        alloc.allocate("clawker.t.y", 1, explicit=[5])


def test_reserved_indices_never_allocated(isolated_env):
    db = StateDB()
    try:
        alloc = GPUAllocator(db, inventory=_inv(4), reserve=[0, 1])
        assert alloc.free_indices() == [2, 3]
        got = alloc.allocate("clawker.t.r", 2)
        assert set(got) == {2, 3}
        with pytest.raises(GPUAllocationError):
            alloc.allocate("clawker.t.r2", 1)
    finally:
        db.close()


def test_stale_reclaim(alloc):
    alloc.allocate("clawker.t.gone", 3)
    freed = alloc.reclaim_stale(live_sandboxes={"clawker.t.other"})
    assert len(freed) == 3
    assert len(alloc.free_indices()) == 8


def test_xgmi_adjacent_preference(isolated_env):
    """Partial adjacency: pick a mutually-linked set when possible."""
    db = StateDB()
    try:
        devs = [GPUDevice(index=i, render_minor=128 + i, card=i) for i in range(4)]
        # two xGMI islands: {0,1} and {2,3}
        devs[0].xgmi_peers = [1]
        devs[1].xgmi_peers = [0]
        devs[2].xgmi_peers = [3]
        devs[3].xgmi_peers = [2]
        inv = GPUInventory(devs)
        alloc = GPUAllocator(db, inventory=inv)
        alloc.allocate("clawker.t.hold", 1, explicit=[0])
        # requesting 2: [1] has no free peer; [2,3] is the adjacent set
        got = alloc.allocate("clawker.t.pair", 2)
        assert set(got) == {2, 3}
    finally:
        db.close()


def test_hbm_budget_env(isolated_env, tmp_path):
    """HBM budget surfaces as CLAWKER_HBM_GB + GPU_MAX_ALLOC_PERCENT."""
    import json
    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    ws = tmp_path / "hbm"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text(
        "project: hbm\nagent:\n  harness: echo\ngpu:\n  count: 1\n  hbm_gb: 72\n")
    orch = Orchestrator(load_config(ws))
    orch.allocator.inventory = _inv(1)
    try:
        info = orch.create(RunOptions(agent="a", name="clawker.hbm.a",
                                      cmd=["true"], firewall=False))
        spec = json.loads((info.rundir / "spec.json").read_text())
        assert spec["env"]["CLAWKER_HBM_GB"] == "72"
        assert spec["env"]["GPU_MAX_ALLOC_PERCENT"] == "25"   # 72/288
        assert spec["env"]["CLAWKER_GPU"] == "0"
        assert {"path": "/dev/kfd"} in spec["devices"]
    finally:
        for i in orch.engine.list():
            orch.teardown(i.name, force=True)
        orch.close()
