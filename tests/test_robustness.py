"""Restart, concurrency and failure-path robustness (reference: CP
resilience contract + ReapFailedStart + reconcile behaviors)."""
import json
import subprocess
import sys
import threading
import time
from pathlib import Path

import pytest

from conftest import requires_isolation

REPO = Path(__file__).resolve().parent.parent


@pytest.fixture
def orch(isolated_env, tmp_path):
    ws = tmp_path / "rproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: rtest\n")
    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator
    o = Orchestrator(load_config(ws))
    yield o
    for info in o.engine.list():
        try:
            o.teardown(info.name, force=True)
        except Exception:
            pass
    o.close()


@requires_isolation
def test_sandbox_restart_cycle(orch):
    """stop -> start -> stop again: run state cleared, init marker
    persists (one-time InitPlan semantics across restarts)."""
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.rtest.re"
    opts = RunOptions(agent="re", name=name, cmd=["/bin/sh", "-c", "echo run-$$; sleep 30"])
    opts.autostart = False
    orch.run(opts)
    with orch.client(name) as c:
        c.agent_initialized()
        c.agent_ready()
    orch.engine.stop(name)
    assert orch.engine.inspect(name).state == "exited"
    # restart the SAME sandbox
    orch.engine.start(name)
    with orch.client(name) as c:
        h = c.hello()
        assert h["initialized"] is True     # marker survived the restart
        assert h["cmd_running"] is False
        c.agent_ready()
    info = orch.engine.inspect(name)
    assert info.state == "running"
    code = orch.engine.stop(name)
    assert code is not None


@requires_isolation
def test_concurrent_exec_clients(orch):
    """Multiple control connections exec simultaneously (ckd poll loop
    multiplexing)."""
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.rtest.cc"
    orch.run(RunOptions(agent="cc", name=name, autostart=True,
                        cmd=["sleep", "30"]))
    results = []
    errors = []

    def worker(i):
        try:
            code, out, _ = orch.engine.exec(
                name, ["/bin/sh", "-c", f"echo w{i}; sleep 0.{i % 3}; echo d{i}"])
            results.append((i, code, out))
        except Exception as e:
            errors.append((i, e))

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(6)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=30)
    assert not errors, errors
    assert len(results) == 6
    for i, code, out in results:
        assert code == 0
        assert f"w{i}".encode() in out and f"d{i}".encode() in out
    orch.engine.stop(name)


@requires_isolation
def test_failed_start_reclaims_gpu_allocations(orch, monkeypatch):
    """create-scope reclaim: a failing start releases the sandbox's GPUs
    (reference: createScope + ReapFailedStart)."""
    from clawker_amd.gpu.inventory import GPUDevice, GPUInventory
    from clawker_amd.orchestrator import RunOptions
    # fake a GPU so allocation happens on this CPU host
    fake = GPUInventory([GPUDevice(index=0, render_minor=128, card=0)])
    orch.allocator.inventory = fake
    # sabotage the runtime so start fails
    monkeypatch.setenv("CLAWKER_NATIVE_BIN", "/nonexistent")
    with pytest.raises(Exception):
        orch.run(RunOptions(agent="fail", name="clawker.rtest.fail",
                            gpus=1, cmd=["true"]))
    assert orch.allocator.allocations() == {}
    assert orch.engine.db.get_sandbox("clawker.rtest.fail") is None


@requires_isolation
def test_cp_survives_subsystem_errors(orch, isolated_env):
    """cpd keeps serving after an op raises (no-exit-after-ready)."""
    from clawker_amd.controlplane.client import CPClient
    cp = CPClient()
    cp.ensure_running()
    try:
        # bad op payload -> error response, daemon stays up
        with pytest.raises(Exception):
            cp.request({"op": "fw_attach"})   # missing fields
        assert cp.status()["ready"] is True
        with pytest.raises(Exception):
            cp.request({"op": "nonsense"})
        assert cp.status()["ready"] is True
    finally:
        cp.stop()


@requires_isolation
def test_stale_state_cleared_on_restart_after_kill(orch):
    """SIGKILL the shim+sandbox, then start again cleanly."""
    import os
    import signal as sig
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.rtest.k9"
    orch.run(RunOptions(agent="k9", name=name, autostart=True, cmd=["sleep", "60"]))
    info = orch.engine.inspect(name)
    os.kill(info.pid, sig.SIGKILL)
    deadline = time.monotonic() + 10
    while time.monotonic() < deadline:
        if orch.engine.inspect(name).state != "running":
            break
        time.sleep(0.05)
    assert orch.engine.inspect(name).state in ("exited", "dead")
    orch.engine.start(name)
    assert orch.engine.inspect(name).state == "running"
    orch.engine.stop(name)


@requires_isolation
def test_restart_policy_on_failure(orch):
    """on-failure:N restart policy (reference: the CP's on-failure:3)."""
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.rtest.rp"
    # fails twice (marker counts attempts), succeeds on the third
    # user="root": /run/clawker is the only rundir path that survives a
    # restart (fresh mount ns each attempt) and it is root-only by design
    opts = RunOptions(
        agent="rp", name=name, autostart=True, restart="on-failure:3",
        user="root",
        cmd=["/bin/sh", "-c",
             "n=$(cat /run/clawker/attempts 2>/dev/null || echo 0); "
             "n=$((n+1)); echo $n > /run/clawker/attempts; "
             "echo attempt-$n; [ $n -ge 3 ] && exit 0 || exit 7"])
    orch.run(opts)
    code = orch.engine.wait(name, timeout_s=60)
    logs = orch.engine.logs(name).decode()
    assert code == 0, logs
    assert "attempt-1" in logs and "attempt-3" in logs
    orch.teardown(name, force=True)

    # exhausted retries: final failure code propagates
    name2 = "clawker.rtest.rp2"
    orch.run(RunOptions(agent="rp2", name=name2, autostart=True,
                        restart="on-failure:2", cmd=["/bin/sh", "-c", "exit 5"]))
    assert orch.engine.wait(name2, timeout_s=60) == 5
    orch.teardown(name2, force=True)


@requires_isolation
def test_ckd_survives_garbage_on_control_socket(orch):
    """Malformed frames / random bytes on ctl.sock must not kill PID 1."""
    import socket as _socket
    import struct
    from clawker_amd.engine import wire
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.rtest.fz"
    orch.run(RunOptions(agent="fz", name=name, autostart=True, cmd=["sleep", "30"]))
    sock_path = orch.engine.ctl_sock(name)
    # 1. raw garbage
    s = wire.connect_unix(sock_path, timeout=5)
    s.sendall(b"\x00\x00\x00\x05notjs")
    s.close()
    # 2. huge length prefix
    s = wire.connect_unix(sock_path, timeout=5)
    s.sendall(struct.pack(">I", 0x7FFFFFFF))
    s.close()
    # 3. valid frame, unknown command + missing fields
    s = wire.connect_unix(sock_path, timeout=5)
    wire.send_frame(s, {"t": "exec"})            # no id/stages
    wire.send_frame(s, {"t": "wat"})
    r = wire.recv_frame(s)
    assert r and r.get("t") == "error"
    s.close()
    # 4. partial frame then disconnect
    s = wire.connect_unix(sock_path, timeout=5)
    s.sendall(b"\x00\x00")
    s.close()
    # ckd still alive and serving
    with orch.client(name) as c:
        h = c.hello()
        assert h["cmd_running"] is True
    code, out, _ = orch.engine.exec(name, ["/bin/echo", "alive"])
    assert code == 0 and b"alive" in out
    orch.engine.stop(name)


@requires_isolation
def test_audit_events_written(orch):
    """ckd writes the load-bearing audit lines (reference: clawkerd
    session/shell_command audit contract)."""
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.rtest.au"
    orch.run(RunOptions(agent="au", name=name, autostart=True,
                        cmd=["/bin/sh", "-c", "sleep 2; exit 4"]))
    orch.engine.exec(name, ["/bin/echo", "x"])
    orch.engine.wait(name, timeout_s=30)
    audit = (orch.engine.inspect(name).rundir / "audit.jsonl").read_text()
    events = [json.loads(l)["event"] for l in audit.splitlines()]
    assert "agent_spawned" in events
    assert "session_started" in events
    assert "shell_command_started" in events
    assert "shell_command_done" in events
    assert "agent_exit" in events


@requires_isolation
def test_concurrent_start_race_single_instance(orch):
    """Two racing starts yield exactly one running instance."""
    from clawker_amd.errors import ConflictError
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.rtest.race"
    opts = RunOptions(agent="race", name=name, cmd=["sleep", "20"])
    opts.autostart = True
    orch.create(opts)
    results = []

    def starter():
        try:
            orch.engine.start(name)
            results.append("started")
        except ConflictError:
            results.append("conflict")
        except Exception as e:
            results.append(f"error:{e}")

    ts = [threading.Thread(target=starter) for _ in range(3)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=30)
    assert results.count("started") == 1, results
    assert all(r in ("started", "conflict") for r in results), results
    # exactly one ckd instance
    code, out, _ = orch.engine.exec(name, ["/bin/sh", "-c", "echo one"])
    assert code == 0
    orch.engine.stop(name)


@requires_isolation
def test_multiple_attach_clients_see_console(orch):
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.rtest.ma"
    opts = RunOptions(agent="ma", name=name,
                      cmd=["/bin/sh", "-c", "sleep 0.3; echo SHARED-LINE; sleep 1"])
    opts.autostart = False
    orch.run(opts)
    c1 = orch.client(name)
    c2 = orch.client(name)
    try:
        c1.attach()
        c2.attach()
        c1.agent_ready()
        seen = []
        for c in (c1, c2):
            data = b""
            for fr in c.stream_events():
                if fr.get("t") == "console":
                    from clawker_amd.engine import wire
                    data += wire.unb64(fr.get("data", ""))
                if b"SHARED-LINE" in data or fr.get("t") == "agent_exit":
                    break
            seen.append(b"SHARED-LINE" in data)
        assert seen == [True, True]
    finally:
        c1.close()
        c2.close()
        orch.engine.stop(name)


@requires_isolation
def test_ckd_survives_malformed_frames(isolated_env):
    """PID-1 resilience: garbage and hostile frames never take ckd down
    (reference: panic-recovery-everywhere contract, recover.go:32)."""
    import socket
    import struct
    from clawker_amd import consts
    from clawker_amd.engine import Engine, SandboxSpec
    from clawker_amd.engine import wire
    eng = Engine()
    name = consts.SANDBOX_NAME_PREFIX + "rb.frames"
    eng.create(SandboxSpec(name=name, hostname="x", autostart=True, netns=True,
                           cmd=["/bin/sleep", "30"]))
    eng.start(name)
    sock_path = eng.ctl_sock(name)
    try:
        # 1: valid frame with invalid JSON -> error frame, conn usable
        s = wire.connect_unix(sock_path)
        body = b"{not json"
        s.sendall(struct.pack(">I", len(body)) + body)
        resp = wire.recv_frame(s)
        assert resp["t"] == "error"
        wire.send_frame(s, {"t": "hello"})
        assert wire.recv_frame(s)["t"] == "hello"
        s.close()
        # 2: oversized length prefix -> dropped, daemon alive
        s = wire.connect_unix(sock_path)
        s.sendall(struct.pack(">I", 1 << 30))
        assert wire.recv_frame(s) is None      # ckd closed us
        s.close()
        # 3: raw garbage (no framing) -> dropped eventually, daemon alive
        s = wire.connect_unix(sock_path)
        s.sendall(b"\xff" * 64)
        s.close()
        # daemon still fully functional
        code, out, _ = eng.exec(name, ["/bin/echo", "alive"])
        assert code == 0 and b"alive" in out
    finally:
        eng.remove(name, force=True)
        eng.close()


@requires_isolation
def test_devbpf_cgroup_device_enforcement():
    """cgroup-v2 device allow-list (VERDICT r01 #3 done-criterion): a
    device NOT in the allow-list cannot be opened from inside the
    enforced cgroup even though the node exists. Skips when the host
    refuses bpf() (restricted CI) or runs cgroup v1."""
    import json
    import subprocess
    from pathlib import Path
    # pure-v2 root, or the hybrid host's unified mount — BPF device
    # programs attach to any cgroup2 directory
    if Path("/sys/fs/cgroup/cgroup.controllers").exists():
        roots = [Path("/sys/fs/cgroup")]
    elif Path("/sys/fs/cgroup/unified/cgroup.controllers").exists():
        roots = [Path("/sys/fs/cgroup/unified")]
    else:
        pytest.skip("no cgroup2 hierarchy on this host")
    base = None
    for root in roots:
        cand = root / "clawker-test"
        try:
            cand.mkdir(exist_ok=True)
            base = cand
            break
        except OSError:
            try:
                own = Path("/proc/self/cgroup").read_text().split(
                    "0::")[1].strip()
                cand = Path(str(root) + own) / "clawker-test"
                cand.mkdir(parents=True, exist_ok=True)
                base = cand
                break
            except (OSError, IndexError):
                continue
    if base is None:
        pytest.skip("no writable cgroup2 subtree")
    probe = Path(__file__).resolve().parents[1] / "native/bin/devbpf_probe"
    r = subprocess.run([str(probe), str(base)], capture_output=True,
                       text=True, timeout=30)
    try:
        base.rmdir()
    except OSError:
        pass
    doc = json.loads(r.stdout.strip() or "{}")
    if r.returncode == 3:
        pytest.skip(f"bpf unavailable on this host: {doc.get('msg')}")
    assert r.returncode == 0, r.stdout + r.stderr
    assert doc["status"] == "enforced"


@requires_isolation
def test_readonly_bind_covers_submounts(orch, tmp_path):
    """a ro passthrough bind must also lock nested
    host submounts (mount_setattr AT_RECURSIVE in ckrt bind_file)."""
    import subprocess
    host_dir = tmp_path / "rotree"
    sub = host_dir / "sub"
    sub.mkdir(parents=True)
    r = subprocess.run(["mount", "-t", "tmpfs", "tmpfs", str(sub)],
                       capture_output=True, text=True)
    if r.returncode != 0:
        pytest.skip(f"cannot create submount: {r.stderr.strip()}")
    try:
        (host_dir / "top.txt").write_text("t")
        (sub / "inner.txt").write_text("i")
        from clawker_amd.engine.spec import Mount
        from clawker_amd.orchestrator import RunOptions
        name = "clawker.rtest.rosub"
        orch.run(RunOptions(
            agent="rosub", name=name, autostart=True, user="root",
            mounts=[Mount(src=str(host_dir), dst="/mnt/rotree", ro=True)],
            cmd=["/bin/sh", "-c",
                 "cat /mnt/rotree/sub/inner.txt; "
                 "touch /mnt/rotree/x 2>&1; touch /mnt/rotree/sub/x 2>&1; "
                 "echo DONE"]))
        code = orch.engine.wait(name, timeout_s=30)
        logs = orch.engine.logs(name).decode()
        assert code == 0, logs
        assert "i" in logs                      # submount content visible
        assert logs.count("Read-only file system") == 2, logs
        orch.teardown(name, force=True)
    finally:
        subprocess.run(["umount", str(sub)], capture_output=True)


@requires_isolation
def test_console_log_rotation(orch):
    """A flooding agent must not grow console.log unboundedly (64 MiB
    rotation to console.log.1 — the docker/lumberjack analog)."""
    from clawker_amd.orchestrator import RunOptions
    name = "clawker.rtest.flood"
    # ~80 MiB of output: forces one rotation
    orch.run(RunOptions(
        agent="flood", name=name, autostart=True, user="root",
        cmd=["/bin/sh", "-c",
             "i=0; while [ $i -lt 80 ]; do head -c 1048576 /dev/zero |"
             " tr '\\0' 'x'; i=$((i+1)); done; echo FLOOD_DONE"]))
    code = orch.engine.wait(name, timeout_s=120)
    assert code == 0
    rundir = orch.engine.inspect(name).rundir
    main = (rundir / "console.log").stat().st_size
    rotated = (rundir / "console.log.1")
    assert rotated.exists(), "no rotation happened"
    assert main < 70 * 1024 * 1024, f"console.log grew to {main}"
    logs = orch.engine.logs(name).decode(errors="replace")
    assert "FLOOD_DONE" in logs
    orch.teardown(name, force=True)
