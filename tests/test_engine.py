"""Engine lifecycle tests with REAL sandboxes (we run as root with
namespace support in CI; reference analog: test/e2e against real dockerd,
but our runtime needs no daemon so these are plain unit tests)."""
import os
import subprocess
import time
from pathlib import Path

import pytest

from clawker_amd import consts
from clawker_amd.engine import Engine, Mount, SandboxSpec
from clawker_amd.engine.images import HOSTFS, ImageStore
from clawker_amd.engine.state import StateDB
from clawker_amd.errors import ConflictError, NotFoundError

from conftest import requires_isolation


# native binaries are built once per session by the conftest fixture


@pytest.fixture
def engine(isolated_env):
    eng = Engine()
    yield eng
    for info in eng.list():
        try:
            eng.remove(info.name, force=True)
        except Exception:
            pass
    eng.close()


def _spec(name, cmd, **kw):
    return SandboxSpec(
        name=f"{consts.SANDBOX_NAME_PREFIX}test.{name}",
        hostname="sbx", autostart=True, netns=True,
        cmd=["/bin/sh", "-c", cmd], **kw)


@requires_isolation
def test_echo_lifecycle(engine):
    info = engine.create(_spec("echo", "echo hello-$FOO; exit 3", env={"FOO": "bar"}))
    assert info.state == "created"
    engine.start(info.name)
    code = engine.wait(info.name, timeout_s=15)
    assert code == 3
    assert b"hello-bar" in engine.logs(info.name)
    info = engine.inspect(info.name)
    assert info.state == "exited" and info.exit_code == 3
    engine.remove(info.name)
    with pytest.raises(NotFoundError):
        engine.inspect(info.name)


@requires_isolation
def test_isolation_properties(engine):
    info = engine.create(_spec(
        "iso", "hostname; echo pid1=$$; ls /dev | wc -l; cat /proc/sys/kernel/hostname"))
    engine.start(info.name)
    assert engine.wait(info.name, timeout_s=15) == 0
    out = engine.logs(info.name).decode()
    assert "sbx" in out            # UTS namespace
    # fresh PID namespace: ckd is PID 1, the agent shell gets a tiny pid
    pid = int(out.split("pid1=")[1].split()[0])
    assert pid < 10


@requires_isolation
def test_exec_into_running_sandbox(engine):
    info = engine.create(_spec("exec", "sleep 30"))
    engine.start(info.name)
    code, out, err = engine.exec(info.name, ["/bin/sh", "-c", "echo from-exec; id -u"])
    assert code == 0
    assert b"from-exec" in out
    # pipeline staging: stage 2 consumes stage 1 stdout
    with engine.client(info.name) as c:
        code, out, _ = c.exec([
            {"argv": ["/bin/echo", "a b c"]},
            {"argv": ["/usr/bin/tr", "a-z", "A-Z"]},
        ])
    assert code == 0 and b"A B C" in out
    engine.stop(info.name)
    assert engine.inspect(info.name).state == "exited"


@requires_isolation
def test_stop_forwards_sigterm(engine):
    info = engine.create(_spec(
        "term", "trap 'exit 42' TERM; echo up; while true; do sleep 0.1; done"))
    engine.start(info.name)
    # wait for the trap to be installed
    deadline = time.monotonic() + 10
    while b"up" not in engine.logs(info.name) and time.monotonic() < deadline:
        time.sleep(0.02)
    code = engine.stop(info.name)
    assert code == 42


@requires_isolation
def test_writable_layer_isolated_from_host(engine, tmp_path):
    info = engine.create(_spec("cow", "echo data > /cow-test-file && cat /cow-test-file"))
    engine.start(info.name)
    assert engine.wait(info.name) == 0
    assert not Path("/cow-test-file").exists()      # COW upper, not host /
    upper = info.statedir / "upper" / "cow-test-file"
    assert upper.read_text().strip() == "data"


@requires_isolation
def test_bind_mount_workspace(engine, tmp_path):
    ws = tmp_path / "ws"
    ws.mkdir()
    (ws / "input.txt").write_text("payload")
    from clawker_amd.engine.spec import Mount
    spec = _spec("ws", "cat /workspace/input.txt; echo done >> /workspace/out.txt",
                 mounts=[Mount(src=str(ws), dst="/workspace")], workdir="/workspace")
    info = engine.create(spec)
    engine.start(info.name)
    assert engine.wait(info.name) == 0
    assert b"payload" in engine.logs(info.name)
    assert (ws / "out.txt").read_text().strip() == "done"   # rw bind


@requires_isolation
def test_netns_denies_egress_by_construction(engine):
    # 203.0.113.1 is TEST-NET; with an uplink-less netns any connect fails fast
    info = engine.create(_spec(
        "net", "ls /sys/class/net; python3 -c \"import socket;s=socket.socket();s.settimeout(2);\n"
               "import sys\ntry: s.connect(('203.0.113.1',80)); sys.exit(1)\n"
               "except OSError: sys.exit(0)\""))
    engine.start(info.name)
    assert engine.wait(info.name, timeout_s=20) == 0
    out = engine.logs(info.name).decode()
    assert "eth0" not in out     # no uplink interface at all


@requires_isolation
def test_agent_ready_gate(engine):
    """autostart=False: CMD runs only after the CP sends agent_ready
    (reference: clawkerd AgentReady releasing the user CMD)."""
    spec = _spec("gate", "echo released")
    spec.autostart = False
    info = engine.create(spec)
    engine.start(info.name)
    with engine.client(info.name) as c:
        h = c.hello()
        assert h["cmd_running"] is False
        assert h["initialized"] is False
        c.agent_initialized()
        pid = c.agent_ready()
        assert pid > 0
    assert engine.wait(info.name) == 0
    assert b"released" in engine.logs(info.name)
    # init marker persisted in the upper layer
    marker = info.statedir / "upper" / "var/lib/clawker/initialized"
    assert marker.exists()
    # ...and survives a restart: hello reports initialized=True so the
    # CP's InitPlan runs exactly once per sandbox (reference: clawkerd
    # Hello handler + AgentInitialized marker, session.go:421)
    engine.start(info.name)
    with engine.client(info.name) as c:
        assert c.hello()["initialized"] is True
        c.agent_ready()
    engine.wait(info.name)


def test_duplicate_create_conflicts(engine):
    if not os.path.exists("/proc"):
        pytest.skip("linux only")
    s1 = _spec("dup", "true")
    engine.create(s1)
    with pytest.raises(ConflictError):
        engine.create(_spec("dup", "true"))


def test_unmanaged_name_rejected(engine):
    from clawker_amd.errors import EngineError
    with pytest.raises(EngineError):
        engine.create(SandboxSpec(name="rogue", cmd=["true"]))


def test_volumes(engine):
    p, fresh = engine.ensure_volume("clawker.test.vol")
    assert fresh and p.is_dir()
    p2, fresh2 = engine.ensure_volume("clawker.test.vol")
    assert p2 == p and not fresh2
    engine.remove_volume("clawker.test.vol")
    assert engine.db.get_volume("clawker.test.vol") is None


def test_image_store_layers(isolated_env, tmp_path):
    store = ImageStore()
    lid, fs = store.new_layer_dir()
    (fs / "etc").mkdir()
    (fs / "etc" / "marker").write_text("layer1")
    final = store.commit_layer(lid)
    from clawker_amd.engine.images import ImageMeta
    store.put(ImageMeta(name="clawker-t:base", layers=[final]))
    lowers = store.lowerdirs_for("clawker-t:base")
    assert lowers[-1] == "/"               # hostfs at the bottom
    assert str(store.layer_path(final)) == lowers[0]
    # second image derived from the first
    lid2, fs2 = store.new_layer_dir()
    (fs2 / "top").write_text("2")
    final2 = store.commit_layer(lid2)
    store.put(ImageMeta(name="clawker-t:h", layers=[final2], parent="clawker-t:base"))
    lowers2 = store.lowerdirs_for("clawker-t:h")
    assert [Path(l).parent.name for l in lowers2[:2]] == [final2, final]
    # prune: removing the child keeps shared layers of the parent
    store.remove("clawker-t:h")
    assert store.layer_path(final)         # still there
    with pytest.raises(Exception):
        store.layer_path(final2)


@requires_isolation
def test_pause_unpause_freezes_workload(engine, tmp_path):
    """pause SIGSTOPs the tree under ckd (counter stops advancing);
    unpause resumes it; ps state surfaces 'paused'; stop still works."""
    import time as _t
    out = tmp_path / "count"
    out.touch()
    info = engine.create(_spec(
        "pz", f"i=0; while true; do i=$((i+1)); echo $i > {out}; "
              "sleep 0.05; done",
        mounts=[Mount(src=str(tmp_path), dst=str(tmp_path), ro=False)]))
    engine.start(info.name)
    deadline = _t.monotonic() + 10
    while out.read_text().strip() in ("", "0") and _t.monotonic() < deadline:
        _t.sleep(0.02)
    assert engine.pause(info.name) >= 1
    assert engine.inspect(info.name).state == "paused"
    v1 = out.read_text()
    _t.sleep(0.4)
    assert out.read_text() == v1          # frozen: no progress
    engine.unpause(info.name)
    assert engine.inspect(info.name).state == "running"
    deadline = _t.monotonic() + 10
    while out.read_text() == v1 and _t.monotonic() < deadline:
        _t.sleep(0.02)
    assert out.read_text() != v1          # thawed: progress resumes
    engine.stop(info.name, timeout_s=5)
    assert engine.inspect(info.name).state == "exited"
    # restart clears the paused marker
    engine.start(info.name)
    assert engine.inspect(info.name).state == "running"
    engine.remove(info.name, force=True)


@requires_isolation
def test_stop_thaws_paused_sandbox(engine):
    """stop on a paused sandbox unfreezes then terminates (no stranded
    SIGSTOPped processes)."""
    info = engine.create(_spec("pzstop", "trap 'exit 7' TERM; while true; do sleep 0.1; done"))
    engine.start(info.name)
    time.sleep(0.3)
    engine.pause(info.name)
    assert engine.inspect(info.name).state == "paused"
    code = engine.stop(info.name, timeout_s=8)
    assert code == 7, code         # the TERM trap ran -> tree was thawed
    assert engine.inspect(info.name).state == "exited"
    engine.remove(info.name, force=True)


@requires_isolation
def test_exec_outlives_socket_timeout(engine):
    """A command longer than the client's socket timeout must complete:
    the timeout guards connect/handshake, not exec duration."""
    info = engine.create(_spec("slowexec", "sleep 60"))
    engine.start(info.name)
    with engine.client(info.name, timeout=1.0) as c:
        t0 = time.monotonic()
        code, out, _ = c.exec([{"argv": ["/bin/sh", "-c",
                                         "sleep 2.5; echo long-ok"]}])
        assert code == 0 and b"long-ok" in out
        assert time.monotonic() - t0 >= 2.4
    engine.remove(info.name, force=True)


@requires_isolation
def test_exec_storm_32_concurrent_clients(engine):
    """32 concurrent exec clients against ONE sandbox's ctl.sock: ckd
    must serve them all (multi-client frame loop + 128-deep accept
    backlog — the r02 connect-storm bug class, validated at the ckd
    tier; fleet tools fan execs out like this)."""
    import threading

    info = engine.create(_spec("xstorm", "sleep 60"))
    engine.start(info.name)
    results: list[tuple[int, bytes]] = []
    errors: list[str] = []
    lock = threading.Lock()

    def one(i: int) -> None:
        try:
            code, out, _ = engine.exec(
                info.name, ["/bin/sh", "-c", f"echo exec-{i}"])
            with lock:
                results.append((code, out))
        except Exception as e:  # noqa: BLE001
            with lock:
                errors.append(f"{i}: {e}")

    ts = [threading.Thread(target=one, args=(i,)) for i in range(32)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=60)
    assert not errors, errors[:5]
    assert len(results) == 32
    assert all(code == 0 for code, _ in results)
    outs = b"".join(o for _, o in results)
    for i in range(32):
        assert f"exec-{i}".encode() in outs
    engine.stop(info.name)
