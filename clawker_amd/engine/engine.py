"""The sandbox engine: create/start/stop/remove/exec/logs over the native
runtime (ckrt/ckd).

Reference analog: pkg/whail (Docker SDK wrapper with managed-label
isolation, engine.go:40) + internal/docker middleware. Here there is no
dockerd: the engine drives ckrt directly, so "managed isolation" is
structural — the engine only ever sees sandboxes in its own state DB and
store directories.
"""
from __future__ import annotations

import os
import shutil
import signal
import subprocess
import time
from dataclasses import dataclass
from pathlib import Path

from .. import consts
from ..errors import ConflictError, EngineError, NotFoundError
from ..logger import get as get_logger
from .ckd_client import CkdClient
from .images import HOSTFS, ImageStore
from .spec import Mount, SandboxSpec
from .state import StateDB
from .users import is_named_user, materialize_user

log = get_logger("engine")

# virtual filesystems never passed through into sandboxes
_VIRTUAL_FSTYPES = {
    "proc", "sysfs", "devtmpfs", "devpts", "tmpfs", "cgroup", "cgroup2",
    "mqueue", "debugfs", "tracefs", "securityfs", "pstore", "bpf",
    "hugetlbfs", "configfs", "fusectl", "binfmt_misc", "autofs", "overlay",
    "squashfs", "ramfs", "rpc_pipefs", "nsfs", "efivarfs",
}
_EXCLUDED_PREFIXES = ("/proc", "/sys", "/dev", "/run", "/tmp", "/boot",
                      "/var/lib/docker", "/var/run")


def native_bin_dir() -> Path:
    v = os.environ.get("CLAWKER_NATIVE_BIN")
    if v:
        return Path(v)
    return Path(__file__).resolve().parents[2] / "native" / "bin"


_backend_cache: str | None = None


def detect_backend() -> str:
    """Probe the strongest available isolation backend.

    ns  — full namespaces + overlayfs + pivot_root (rootful node with
          CAP_SYS_ADMIN: the production MI355X deployment target).
    proc — plain supervised child processes (hosts that forbid namespace
          creation, e.g. restricted CI containers with
          user.max_user_namespaces=0): env-contract + cgroup-best-effort +
          ROCR_VISIBLE_DEVICES GPU selection, no mount/net isolation.

    Override with CLAWKER_BACKEND=ns|proc.
    """
    global _backend_cache
    v = os.environ.get("CLAWKER_BACKEND")
    if v in ("ns", "proc"):
        return v
    if _backend_cache is None:
        probe = backend_probe()
        _backend_cache = probe["backend"]
        if _backend_cache == "proc":
            log.warn("isolation_degraded", backend="proc",
                     reason=probe.get("ns_error", "namespace creation unavailable"))
    return _backend_cache


_probe_cache: dict | None = None


def backend_probe() -> dict:
    """Machine-readable isolation probe: attempts namespace creation and
    an overlay mount inside the fresh ns, recording the exact failure so
    degraded bench records can say WHY (the GPU lease
    forbids namespaces — the record must carry the errno, not a silently
    optimistic flag)."""
    global _probe_cache
    if _probe_cache is not None:
        return _probe_cache
    out: dict = {"backend": "proc"}
    try:
        r = subprocess.run(["unshare", "-pmf", "true"],
                           capture_output=True, text=True, timeout=10)
    except (OSError, subprocess.TimeoutExpired) as e:
        out["ns_error"] = f"unshare probe: {e}"
        _probe_cache = out
        return out
    if r.returncode != 0:
        out["ns_error"] = (r.stderr or "").strip() or f"unshare rc={r.returncode}"
        _probe_cache = out
        return out
    # namespaces work; can we overlay-mount inside one?
    import tempfile
    with tempfile.TemporaryDirectory() as td:
        for sub in ("up", "work", "m"):
            os.makedirs(os.path.join(td, sub))
        r2 = subprocess.run(
            ["unshare", "-mf", "sh", "-c",
             f"mount -t overlay overlay -o "
             f"lowerdir=/etc,upperdir={td}/up,workdir={td}/work {td}/m"],
            capture_output=True, text=True, timeout=10)
    if r2.returncode != 0:
        out["ns_error"] = ("overlay probe: "
                           + ((r2.stderr or "").strip() or f"rc={r2.returncode}"))
        _probe_cache = out
        return out
    out = {"backend": "ns"}
    _probe_cache = out
    return out


def host_passthrough_mounts() -> list[Mount]:
    """Separate real filesystems mounted under / that the hostfs overlay
    lowerdir will NOT traverse (overlayfs ignores lowerdir submounts):
    rebind them read-only so e.g. a separately-mounted /opt/rocm stays
    visible inside sandboxes."""
    out: list[Mount] = []
    seen: set[str] = set()
    try:
        lines = Path("/proc/self/mountinfo").read_text().splitlines()
    except OSError:
        return out
    for line in lines:
        # mountinfo: ... mountpoint ... - fstype source opts
        try:
            left, right = line.split(" - ", 1)
            fields = left.split()
            mountpoint = fields[4]
            fstype = right.split()[0]
        except (ValueError, IndexError):
            continue
        if mountpoint == "/" or fstype in _VIRTUAL_FSTYPES:
            continue
        if any(mountpoint == p or mountpoint.startswith(p + "/") for p in _EXCLUDED_PREFIXES):
            continue
        if any(mountpoint.startswith(s + "/") or mountpoint == s for s in seen):
            continue   # covered by an already-included ancestor (rbind)
        seen.add(mountpoint)
        out.append(Mount(src=mountpoint, dst=mountpoint, ro=True))
    return out


@dataclass
class SandboxInfo:
    name: str
    project: str
    agent: str
    image: str
    created: float
    state: str          # created | running | exited | dead
    exit_code: int | None
    pid: int | None
    gpus: list[int]
    labels: dict
    rundir: Path
    statedir: Path


class Engine:
    def __init__(self, images: ImageStore | None = None, db: StateDB | None = None):
        self.images = images or ImageStore()
        self.db = db or StateDB()
        self.rt_root = consts.runtime_dir() / "sandboxes"
        self.rt_root.mkdir(parents=True, exist_ok=True)
        # 0711: host-side non-root users cannot enumerate sandbox rundirs
        # (each rundir's own modes gate its contents; in-sandbox access
        # goes through the bind mount, unaffected)
        try:
            os.chmod(self.rt_root, 0o711)
        except OSError:
            pass
        self.store_root = consts.sandbox_store_dir()
        self.store_root.mkdir(parents=True, exist_ok=True)
        self._passthrough: list[Mount] | None = None
        self.backend = detect_backend()
        if self.backend == "ns":
            self._ensure_upper_filesystem()

    def _ensure_upper_filesystem(self) -> None:
        """Overlay upperdirs cannot live on an overlayfs. Hosts whose root
        is itself an overlay (CI containers) get a tmpfs mounted over the
        sandbox store; a production node with a real filesystem is
        untouched."""
        try:
            target = self.store_root.resolve()
            best_len = -1
            fstype = ""
            for line in Path("/proc/self/mounts").read_text().splitlines():
                parts = line.split()
                if len(parts) < 3:
                    continue
                mp = parts[1].encode().decode("unicode_escape")
                if (str(target) == mp or str(target).startswith(mp.rstrip("/") + "/")) \
                        and len(mp) > best_len:
                    best_len = len(mp)
                    fstype = parts[2]
            if fstype != "overlay":
                return
            r = subprocess.run(
                ["mount", "-t", "tmpfs", "-o", "mode=700", "clawker-sandboxes",
                 str(self.store_root)], capture_output=True, text=True)
            if r.returncode != 0:
                log.warn("upperfs_unavailable", err=r.stderr.strip())
        except OSError:
            pass

    def close(self) -> None:
        self.db.close()

    # ------------------------------------------------------------- paths ----
    def rundir(self, name: str) -> Path:
        return self.rt_root / name

    def statedir(self, name: str) -> Path:
        return self.store_root / name

    def ctl_sock(self, name: str) -> Path:
        return self.rundir(name) / consts.CKD_SOCK_NAME

    # ------------------------------------------------------------ create ----
    def create(self, spec: SandboxSpec, image: str = HOSTFS) -> SandboxInfo:
        if not spec.name.startswith(consts.SANDBOX_NAME_PREFIX):
            raise EngineError("create", f"unmanaged name: {spec.name}")
        if self.db.get_sandbox(spec.name) is not None:
            raise ConflictError(f"sandbox exists: {spec.name}")
        meta = self.images.get(image)

        rundir = self.rundir(spec.name)
        statedir = self.statedir(spec.name)
        for d in (rundir / "bin", statedir / "upper", statedir / "work"):
            d.mkdir(parents=True, exist_ok=True)
        # 0711: the in-sandbox agent user must traverse /run/clawker (the
        # rundir bind) to reach its bootstrap token, trust bundle, helper
        # bins and the hostproxy/ssh-agent sockets — but must not be able
        # to enumerate it. Root-only material inside carries its own mode
        # (ctl.sock 0600 via ckd, spec.json/policy.json 0600).
        os.chmod(rundir, 0o711)
        os.chmod(rundir / "bin", 0o755)

        # stage the PID-1 supervisor into the rundir (bind-mounted at
        # /run/clawker inside; ckrt execs /run/clawker/bin/ckd)
        for binname in ("ckd", "ckgw"):
            src = native_bin_dir() / binname
            if not src.is_file():
                if binname == "ckd":
                    raise EngineError(
                        "create", f"ckd binary missing: {src} (run `make native`)")
                continue
            dst = rundir / "bin" / binname
            if not dst.exists() or dst.stat().st_mtime < src.stat().st_mtime:
                shutil.copy2(src, dst)
        # host-service helper scripts (hostproxy clients)
        assets = Path(__file__).resolve().parents[1] / "assets"
        for helper in ("host-open.sh", "git-credential-clawker",
                       "clawker-ssh-proxy"):
            src = assets / helper
            if src.is_file():
                shutil.copy2(src, rundir / "bin" / helper)
        # per-agent bootstrap identity material (reference:
        # InstallAgentBootstrapMaterial, agent_bootstrap.go)
        from ..auth import install_bootstrap
        install_bootstrap(rundir, spec.name)
        # MITM trust bundle staged into the rundir when the orchestrator
        # minted one (label carries the host path)
        bundle = spec.labels.pop("dev.clawker.trustbundle", None)
        if bundle and Path(bundle).is_file():
            shutil.copy2(bundle, rundir / "trust-bundle.crt")
            # SSL_CERT_FILE target: the unprivileged agent must read it
            os.chmod(rundir / "trust-bundle.crt", 0o644)


        # rootfs stack from the image + host passthrough binds
        spec.backend = self.backend
        spec.rundir = str(rundir)
        if self.backend == "ns":
            spec.lowerdirs = self.images.lowerdirs_for(image)
            if not spec.upper:        # builds set a staging layer as upper
                spec.upper = str(statedir / "upper")
            if not spec.work:         # workdir must share the upper's fs
                spec.work = str(statedir / "work")
            spec.merged = str(rundir / "merged")
            if self._passthrough is None:
                self._passthrough = host_passthrough_mounts()
            spec.mounts = self._passthrough + spec.mounts
            spec.paths = {"rundir": "/run/clawker",
                          "marker": "/var/lib/clawker/initialized"}
            # per-sandbox identity files written straight into the overlay
            # upper — zero bind mounts for them at boot. Build sandboxes
            # skip this: their upper BECOMES an image layer and must not
            # carry the build container's identity.
            if spec.labels.get("dev.clawker.build") != "true":
                etc = Path(spec.upper) / "etc"
                etc.mkdir(parents=True, exist_ok=True)
                (etc / "hostname").write_text(spec.hostname + "\n")
                (etc / "hosts").write_text(
                    f"127.0.0.1\tlocalhost {spec.hostname}\n::1\tlocalhost\n")
                if spec.netns:
                    # loopback stub resolver (the firewall dnsd path);
                    # without netns the host resolv.conf shows through
                    (etc / "resolv.conf").write_text("nameserver 127.0.0.1\n")
        else:
            # proc backend: ckd runs against the host fs — per-sandbox host
            # paths; no mounts/devices isolation (env contract only)
            spec.mounts = []
            spec.lowerdirs = []
            spec.paths = {"rundir": str(rundir),
                          "marker": str(statedir / "initialized")}

        # image-level env/user/cmd defaults
        env = dict(meta.env)
        env.update(spec.env)
        spec.env = env
        if not spec.user:
            spec.user = meta.user
        if not spec.cmd:
            spec.cmd = list(meta.cmd)
        if spec.workdir in ("", "/") and meta.workdir:
            spec.workdir = meta.workdir

        # named users (`user: agent`) must resolve inside the sandbox;
        # hostfs overlays see the host passwd, so the engine materializes
        # the user into the overlay upper (engine/users.py). ckd refuses
        # to spawn unresolvable named users — never a silent root fall-
        # back — so this is what makes the flagship harness config real.
        if (self.backend == "ns" and is_named_user(spec.user)
                and spec.labels.get("dev.clawker.build") != "true"):
            uid, gid = materialize_user(
                spec.user, Path(spec.upper), spec.lowerdirs,
                uid_hint=spec.uid_hint, gid_hint=spec.gid_hint)
            spec.labels["dev.clawker.uid"] = str(uid)
            spec.labels["dev.clawker.gid"] = str(gid)
            spec.env.setdefault("HOME", f"/home/{spec.user}")

        spec.labels.setdefault(consts.MANAGED_LABEL, "true")
        spec_path = rundir / "spec.json"
        spec.write(spec_path)
        # env may carry secrets (tokens via --env/env_file); root-only
        os.chmod(spec_path, 0o600)

        self.db.add_sandbox(
            spec.name, spec.labels.get(consts.PROJECT_LABEL, ""),
            spec.labels.get(consts.AGENT_LABEL, ""), image, spec.labels,
            [int(x) for x in spec.labels.get(consts.GPU_LABEL, "").split(",") if x != ""],
            str(spec_path), str(rundir), str(statedir))
        log.info("sandbox_created", sandbox=spec.name, image=image)
        return self.inspect(spec.name)

    # ------------------------------------------------------------- start ----
    def start(self, name: str, wait_ready_s: float = 10.0) -> SandboxInfo:
        row = self._row(name)
        rundir = Path(row["rundir"])
        # serialize concurrent starts of the same sandbox (two CLIs racing
        # would otherwise both pass the running check and spawn two shims)
        import fcntl as _fcntl
        lock_f = open(rundir / "start.lock", "w")
        try:
            _fcntl.flock(lock_f, _fcntl.LOCK_EX)
            return self._start_locked(name, row, rundir, wait_ready_s)
        finally:
            _fcntl.flock(lock_f, _fcntl.LOCK_UN)
            lock_f.close()

    def _start_locked(self, name: str, row: dict, rundir: Path,
                      wait_ready_s: float) -> SandboxInfo:
        status = self._status(rundir)
        if status.get("state") == "running" and self._pid_alive(status.get("pid")):
            raise ConflictError(f"sandbox already running: {name}")
        # a dying previous instance's shim may still be reaping; let it
        # finish so its exit.json/status.json writes can't clobber ours
        old_shim = status.get("shim_pid")
        if old_shim and self._pid_alive(int(old_shim)):
            self._wait_pid_gone(int(old_shim), 5.0)
        # clear stale run state (incl. a dead instance's control socket —
        # start() readiness keys on its existence)
        for f in ("exit.json", "status.json", "pid", "console.log", "paused",
                  consts.CKD_SOCK_NAME):
            (rundir / f).unlink(missing_ok=True)

        ckrt = native_bin_dir() / "ckrt"
        if not ckrt.is_file():
            raise EngineError("start", f"ckrt binary missing: {ckrt} (run `make native`)")
        shim_log = open(rundir / "shim.log", "ab")
        proc = subprocess.Popen(
            [str(ckrt), "run", row["spec_path"]],
            stdin=subprocess.DEVNULL, stdout=shim_log, stderr=shim_log,
            start_new_session=True, close_fds=True)
        shim_log.close()

        # wait until ckd binds its control socket (or the sandbox dies fast)
        sock = rundir / consts.CKD_SOCK_NAME
        deadline = time.monotonic() + wait_ready_s
        delay = 0.0003   # backoff: the socket appears ~1-3 ms after fork
        while time.monotonic() < deadline:
            if sock.exists():
                break
            if (rundir / "exit.json").exists() or proc.poll() is not None:
                tail = self._tail(rundir / "shim.log")
                raise EngineError("start", f"sandbox died during boot: {tail}")
            time.sleep(delay)
            delay = min(delay * 2, 0.005)
        else:
            raise EngineError("start", f"timed out waiting for ckd socket ({name})")
        log.info("sandbox_started", sandbox=name, shim_pid=proc.pid)
        return self.inspect(name)

    # -------------------------------------------------------------- stop ----
    def stop(self, name: str, timeout_s: float = 10.0) -> int | None:
        row = self._row(name)
        rundir = Path(row["rundir"])
        if (rundir / "paused").exists():
            # docker semantics: stopping a paused sandbox thaws it first —
            # a SIGTERM against a SIGSTOPped tree would stay pending and
            # the later SIGKILL path can strand frozen processes on the
            # proc backend (no pidns teardown there)
            try:
                self.unpause(name)
            except ConflictError:
                pass
        pid = self._init_pid(rundir)
        if pid is None or not self._pid_alive(pid):
            return self._exit_code(rundir)
        try:
            os.kill(pid, signal.SIGTERM)
        except ProcessLookupError:
            return self._exit_code(rundir)
        deadline = time.monotonic() + timeout_s
        while time.monotonic() < deadline:
            if not self._pid_alive(pid):
                return self._exit_code(rundir)
            time.sleep(0.02)
        try:
            os.kill(pid, signal.SIGKILL)
        except ProcessLookupError:
            pass
        self._wait_pid_gone(pid, 5.0)
        log.info("sandbox_stopped", sandbox=name, forced=True)
        return self._exit_code(rundir)

    def kill(self, name: str, sig: int = signal.SIGKILL) -> None:
        pid = self._init_pid(self.rundir(name))
        if pid is not None and self._pid_alive(pid):
            os.kill(pid, sig)

    # ------------------------------------------------------ pause/unpause --
    @staticmethod
    def _descendants(root_pid: int) -> list[int]:
        """All live descendants of root_pid via /proc children files."""
        out: list[int] = []
        stack = [root_pid]
        while stack:
            pid = stack.pop()
            try:
                for task in Path(f"/proc/{pid}/task").iterdir():
                    kids = (task / "children").read_text().split()
                    for k in kids:
                        out.append(int(k))
                        stack.append(int(k))
            except (OSError, ValueError):
                continue
        return out

    def _signal_workload(self, name: str, sig: int) -> int:
        """Signal every process under the sandbox's init (ckd) — but not
        ckd itself, so the control socket stays responsive while the
        workload is frozen (docker pause freezes PID 1 too; keeping the
        supervisor live is the better trade on this runtime: status and
        unpause keep working)."""
        pid = self._init_pid(self.rundir(name))
        if pid is None or not self._pid_alive(pid):
            raise ConflictError(f"sandbox not running: {name}")
        n = 0
        # repeat until the set is stable: a forking workload can race one
        # sweep, but a STOPPED parent cannot fork again
        for _ in range(10):
            pids = self._descendants(pid)
            sent = 0
            for p in pids:
                try:
                    os.kill(p, sig)
                    sent += 1
                except (ProcessLookupError, PermissionError):
                    pass
            n = max(n, sent)
            if sig != signal.SIGSTOP or not pids:
                break
            state = {p for p in pids
                     if self._proc_state(p) not in ("T", None)}
            if not state:
                break
        return n

    @staticmethod
    def _proc_state(pid: int) -> str | None:
        try:
            return Path(f"/proc/{pid}/stat").read_text().rsplit(")", 1)[1].split()[0]
        except (OSError, IndexError):
            return None

    def pause(self, name: str) -> int:
        """Freeze the sandbox's workload (SIGSTOP to the process tree
        under ckd). Returns number of processes frozen."""
        n = self._signal_workload(name, signal.SIGSTOP)
        (self.rundir(name) / "paused").touch()
        log.info("sandbox_paused", sandbox=name, procs=n)
        return n

    def unpause(self, name: str) -> int:
        n = self._signal_workload(name, signal.SIGCONT)
        try:
            (self.rundir(name) / "paused").unlink()
        except OSError:
            pass
        log.info("sandbox_unpaused", sandbox=name, procs=n)
        return n

    # -------------------------------------------------------------- wait ----
    def wait(self, name: str, timeout_s: float | None = None) -> int:
        row = self._row(name)
        rundir = Path(row["rundir"])
        deadline = None if timeout_s is None else time.monotonic() + timeout_s
        while True:
            code = self._exit_code(rundir)
            if code is not None:
                return code
            pid = self._init_pid(rundir)
            if pid is not None and not self._pid_alive(pid):
                # shim writes exit.json right after reaping; brief grace
                # (a restart-policy shim may instead respawn — detect via a
                # changed pidfile and keep waiting)
                for _ in range(100):
                    code = self._exit_code(rundir)
                    if code is not None:
                        return code
                    new_pid = self._init_pid(rundir)
                    if new_pid not in (None, pid) and self._pid_alive(new_pid):
                        break   # restarted; continue outer wait
                    time.sleep(0.01)
                else:
                    return -1
                continue
            if deadline is not None and time.monotonic() > deadline:
                raise EngineError("wait", f"timeout waiting for {name}")
            time.sleep(0.005)

    # ------------------------------------------------------------ remove ----
    def remove(self, name: str, force: bool = False) -> None:
        row = self._row(name)
        rundir = Path(row["rundir"])
        pid = self._init_pid(rundir)
        if pid is not None and self._pid_alive(pid):
            if not force:
                raise ConflictError(f"sandbox running: {name} (use --force)")
            self.stop(name, timeout_s=3.0)
        # the shim writes exit.json/status.json AFTER reaping ckd — let it
        # finish or its late write re-populates a half-deleted rundir
        # (observed once in a 100k-loop soak)
        shim = self._status(rundir).get("shim_pid")
        if shim and self._pid_alive(int(shim)):
            self._wait_pid_gone(int(shim), 3.0)
        shutil.rmtree(rundir, ignore_errors=True)
        # late-write race residue (shim/ckd tail writes landing mid-rmtree):
        # retry with backoff — a single 50 ms retry still leaked ~1 rundir
        # per 16k loops in the 32-way soak
        delay = 0.05
        while rundir.exists() and delay <= 0.8:
            time.sleep(delay)
            shutil.rmtree(rundir, ignore_errors=True)
            delay *= 2
        if rundir.exists():
            log.warn("rundir_residue", sandbox=name)
        shutil.rmtree(row["statedir"], ignore_errors=True)
        self.db.remove_sandbox(name)
        log.info("sandbox_removed", sandbox=name)

    # ----------------------------------------------------------- inspect ----
    def inspect(self, name: str) -> SandboxInfo:
        row = self._row(name)
        rundir = Path(row["rundir"])
        status = self._status(rundir)
        state = "created"
        pid = None
        exit_code = self._exit_code(rundir)
        if status.get("state") == "running":
            pid = int(status["pid"])
            if self._pid_alive(pid):
                state = "paused" if (rundir / "paused").exists() else "running"
            else:
                state, pid = ("exited", None) if exit_code is not None else ("dead", None)
        elif exit_code is not None:
            state = "exited"
        return SandboxInfo(
            name=name, project=row["project"], agent=row["agent"], image=row["image"],
            created=row["created"], state=state, exit_code=exit_code, pid=pid,
            gpus=row["gpus"], labels=row["labels"], rundir=rundir,
            statedir=Path(row["statedir"]))

    def list(self, project: str | None = None, all_states: bool = True,
             label_filters: dict | None = None) -> list[SandboxInfo]:
        out = []
        for row in self.db.list_sandboxes(project=project, label_filters=label_filters):
            try:
                info = self.inspect(row["name"])
            except NotFoundError:
                # a concurrent remove won the race between the row scan
                # and the inspect — the listing just doesn't include it
                continue
            if not all_states and info.state != "running":
                continue
            out.append(info)
        return out

    # ------------------------------------------------------------- logs  ----
    def logs(self, name: str) -> bytes:
        row = self._row(name)
        p = Path(row["rundir"]) / "console.log"
        return p.read_bytes() if p.exists() else b""

    # ------------------------------------------------------------ client ----
    def client(self, name: str, deadline_s: float = 10.0,
               timeout: float | None = 30.0) -> CkdClient:
        self._row(name)
        return CkdClient.wait_connect(self.ctl_sock(name), deadline_s, timeout=timeout)

    def exec(self, name: str, argv: list[str], user: str = "",
             stdin: bytes = b"", cwd: str = "") -> tuple[int, bytes, bytes]:
        with self.client(name) as c:
            stage: dict = {"argv": argv}
            if user:
                stage["user"] = user
            if cwd:
                stage["cwd"] = cwd
            return c.exec([stage], stdin=stdin)

    # ------------------------------------------------------------- images ---
    def remove_image(self, name: str, force: bool = False) -> None:
        """Remove an image, refusing while any sandbox still references it
        (a running sandbox's overlay must not lose its lowerdirs —
        docker-rmi semantics)."""
        users = [row["name"] for row in self.db.list_sandboxes()
                 if row["image"] == name]
        if users and not force:
            raise ConflictError(
                f"image in use by sandboxes: {users} (remove them or --force)")
        self.images.remove(name)

    # ------------------------------------------------------------ volumes ---
    def ensure_volume(self, name: str, labels: dict | None = None) -> tuple[Path, bool]:
        """Create-or-get a named volume; returns (path, fresh)."""
        row = self.db.get_volume(name)
        if row is not None:
            return Path(row["path"]), False
        path = consts.volume_store_dir() / name
        path.mkdir(parents=True, exist_ok=True)
        self.db.add_volume(name, str(path), labels or {consts.MANAGED_LABEL: "true"})
        return path, True

    def remove_volume(self, name: str) -> None:
        row = self.db.get_volume(name)
        if row is None:
            raise NotFoundError(f"volume not found: {name}")
        shutil.rmtree(row["path"], ignore_errors=True)
        self.db.remove_volume(name)

    # ------------------------------------------------------------ helpers ---
    def _row(self, name: str) -> dict:
        row = self.db.get_sandbox(name)
        if row is None:
            raise NotFoundError(f"sandbox not found: {name}")
        return row

    @staticmethod
    def _status(rundir: Path) -> dict:
        p = rundir / "status.json"
        try:
            import json
            return json.loads(p.read_text())
        except (OSError, ValueError):
            return {}

    @staticmethod
    def _exit_code(rundir: Path) -> int | None:
        p = rundir / "exit.json"
        try:
            import json
            return int(json.loads(p.read_text())["code"])
        except (OSError, ValueError, KeyError):
            return None

    @staticmethod
    def _init_pid(rundir: Path) -> int | None:
        try:
            return int((rundir / "pid").read_text().strip())
        except (OSError, ValueError):
            return None

    @staticmethod
    def _pid_alive(pid: int | None) -> bool:
        if not pid:
            return False
        try:
            os.kill(int(pid), 0)
            return True
        except (ProcessLookupError, PermissionError):
            return False

    @staticmethod
    def _wait_pid_gone(pid: int, timeout_s: float) -> None:
        deadline = time.monotonic() + timeout_s
        while time.monotonic() < deadline:
            try:
                os.kill(pid, 0)
            except ProcessLookupError:
                return
            time.sleep(0.01)

    @staticmethod
    def _tail(path: Path, n: int = 400) -> str:
        try:
            data = path.read_bytes()
            return data[-n:].decode(errors="replace")
        except OSError:
            return ""
