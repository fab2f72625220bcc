"""cpd — the clawker-amd control-plane daemon.

Reference: the clawkercp privileged container (internal/controlplane
cmd.go:367 Main — pub/sub topics, firewall handler + action queue, agent
registry/dialer/executor, healthz, drain-to-zero, SOS). Single-node
redesign: cpd is a host daemon (not a container) owning:

  * the admin API over a root-only Unix socket (framed JSON — the
    AdminService analog; filesystem perms replace mTLS+OAuth2)
  * the agent registry (sqlite; cpd is the SOLE writer — the reference's
    CP-is-sole-sqlite-writer rule)
  * the firewall action queue: a single worker serializing ALL policy
    mutations (reference: firewall/queue.go ActionQueue) — bypass with a
    dead-man timer, rule reloads pushed to running sandbox gateways
  * a sandbox watcher that runs Init/Boot plans for sandboxes started
    detached, and drain-to-zero self-shutdown (reference: AgentWatcher)
  * the event log (jsonl) every subsystem appends to

Resilience contract (reference CLAUDE.md:44-100): after ready, cpd never
exits on subsystem errors — failures degrade per-subsystem with
event=<subsystem>_unavailable logs.
"""
from __future__ import annotations

import json
import os
import queue
import signal
import socket
import sqlite3
import threading
import time
from pathlib import Path

from .. import consts
from ..engine import wire
from ..logger import get as get_logger, setup as logger_setup

log = get_logger("cpd")


def admin_sock_path() -> Path:
    return consts.runtime_dir() / consts.CP_ADMIN_SOCK


def pid_path() -> Path:
    return consts.runtime_dir() / "cpd.pid"


def events_path() -> Path:
    return consts.state_dir() / "cp-events.jsonl"


class EventLog:
    """Append-only jsonl event stream (the pub/sub + OTLP lanes analog)."""

    def __init__(self, path: Path, on_emit=None):
        self.path = path
        self.path.parent.mkdir(parents=True, exist_ok=True)
        self._lock = threading.Lock()
        self.on_emit = on_emit     # live pub/sub fan-out (Topic.publish)

    MAX_BYTES = 10 * 1024 * 1024

    def emit(self, event: str, **kv) -> None:
        rec = {"ts": time.time(), "event": event, **kv}
        with self._lock:
            try:
                if self.path.stat().st_size > self.MAX_BYTES:
                    self.path.replace(self.path.with_suffix(".jsonl.1"))
            except OSError:
                pass
            with open(self.path, "a") as f:
                f.write(json.dumps(rec) + "\n")
        if self.on_emit is not None:
            self.on_emit(rec)

    def tail(self, n: int = 100) -> list[dict]:
        # read a bounded tail window, not the whole (up to 10 MB) file
        try:
            with open(self.path, "rb") as f:
                f.seek(0, 2)
                size = f.tell()
                window = min(size, max(4096, n * 512))
                f.seek(size - window)
                chunk = f.read().decode(errors="replace")
            lines = chunk.splitlines()
            if window < size and lines:
                lines = lines[1:]   # first line may be torn
            out = []
            for l in lines[-n:]:
                if l.strip():
                    try:
                        out.append(json.loads(l))
                    except ValueError:
                        continue
            return out
        except OSError:
            return []


class ActionQueue:
    """Single-goroutine FIFO serializing all firewall mutations
    (reference: firewall/queue.go:122; head-wins coalescing for reloads)."""

    def __init__(self):
        self._q: "queue.Queue[tuple[str, callable]]" = queue.Queue()
        self._worker = threading.Thread(target=self._run, daemon=True)
        self._closed = False
        self._pending_kinds: set[str] = set()
        self._lock = threading.Lock()
        self._worker.start()

    def submit(self, kind: str, fn, coalesce: bool = False) -> None:
        if self._closed:
            return
        with self._lock:
            if coalesce and kind in self._pending_kinds:
                return
            self._pending_kinds.add(kind)
        self._q.put((kind, fn))

    def submit_sync(self, kind: str, fn, timeout: float = 10.0):
        """Run through the queue but wait for completion (rule mutations
        must be visible before the caller proceeds — reference: RuleMutate
        is non-coalescing and callers observe the result)."""
        done = threading.Event()
        box: dict = {}

        def wrapped():
            try:
                box["result"] = fn()
            except Exception as e:
                box["error"] = e
            finally:
                done.set()

        self.submit(kind, wrapped)
        if not done.wait(timeout):
            raise TimeoutError(f"action {kind} timed out")
        if "error" in box:
            raise box["error"]
        return box.get("result")

    def _run(self) -> None:
        while True:
            kind, fn = self._q.get()
            with self._lock:
                self._pending_kinds.discard(kind)
            if fn is None:
                return
            try:
                fn()
            except Exception as e:   # never kill the queue
                log.error("action_failed", kind=kind, err=str(e))

    def close(self) -> None:
        self._closed = True
        self._q.put(("close", None))


class CPDaemon:
    def __init__(self):
        from ..config.config import load_settings
        from ..engine import Engine
        from ..firewall.gateway import GatewayManager
        from .pubsub import Topic
        self.settings = load_settings().get()
        self.engine = Engine()
        self.topic = Topic("cp-events")
        self.events = EventLog(events_path(), on_emit=self.topic.publish)
        self.queue = ActionQueue()
        from ..socketbridge import SocketBridgeManager
        self.gateways = GatewayManager(
            on_event=lambda ev: self.events.emit("egress_decision", **ev),
            event_rate=self.settings.firewall.event_rate_limit,
            event_burst=self.settings.firewall.event_burst,
            dns_upstream=self.settings.firewall.dns_upstream)
        self.bridges = SocketBridgeManager()
        self.ready = False
        self._stop = threading.Event()
        self._bypass_until = 0.0
        self._bypass_timer: threading.Timer | None = None
        self._registry_db = self._open_registry()
        self._last_agent_seen = time.time()
        # sandboxes explicitly firewall-disabled by the admin: the watcher
        # must not re-enroll them (reference: FirewallDisable sticks until
        # FirewallEnable, CLAUDE.md:69 semantics)
        self._fw_disabled: set[str] = set()
        # CP-driven plans for orphaned gated sandboxes (reference:
        # executor model, init_steps.go:67; r01 gap #8: plans were
        # client-coupled — a `run -d` whose CLI died never booted)
        self._orphan_since: dict[str, float] = {}
        self._driving: set[str] = set()
        # HBM budget watchdog (env hints are not
        # enforcement — the CP samples drm fdinfo and kills on breach)
        from ..monitor.hbm import HBMWatchdog
        self.hbm = HBMWatchdog(
            self.engine, self.events,
            mode=getattr(self.settings.gpu, "hbm_enforce", "kill"))

    # ------------------------------------------------------------ registry --
    def _open_registry(self) -> sqlite3.Connection:
        path = consts.data_dir() / "cp-registry.db"
        db = sqlite3.connect(str(path), check_same_thread=False)
        db.execute("PRAGMA journal_mode=WAL")
        db.execute(
            "CREATE TABLE IF NOT EXISTS agents ("
            " sandbox TEXT PRIMARY KEY, project TEXT, agent TEXT,"
            " first_seen REAL, last_seen REAL, state TEXT)")
        db.commit()
        return db

    def _record_agent(self, info) -> None:
        now = time.time()
        self._registry_db.execute(
            "INSERT INTO agents (sandbox,project,agent,first_seen,last_seen,state)"
            " VALUES (?,?,?,?,?,?) ON CONFLICT(sandbox) DO UPDATE SET"
            " last_seen=excluded.last_seen, state=excluded.state",
            (info.name, info.project, info.agent, now, now, info.state))
        self._registry_db.commit()

    def _reconcile_registry(self, infos) -> None:
        """Keep every known sandbox's row current and close out rows whose
        sandbox no longer exists in the engine (so `controlplane agents`
        never shows a removed sandbox as running forever)."""
        for i in infos:
            self._record_agent(i)
        names = {i.name for i in infos}
        self._registry_db.execute(
            "UPDATE agents SET state='removed' WHERE state!='removed' "
            "AND sandbox NOT IN (%s)" % ",".join("?" * len(names))
            if names else
            "UPDATE agents SET state='removed' WHERE state!='removed'",
            tuple(names))
        self._registry_db.commit()

    # ------------------------------------------------------------- watcher --
    def _watch_loop(self) -> None:
        """Track sandbox lifecycle: agent registry, firewall-gateway and
        agent-socket-bridge reconcile, drain-to-zero. (Init/boot plans are
        driven by the starting client — CLI/fleet — which holds the ckd
        session; the watcher only reconciles host-side attachments.)"""
        while not self._stop.is_set():
            running = []
            reconcile_ok = True
            try:
                infos = self.engine.list()
                running = [i for i in infos if i.state in ("running", "paused")]
                self._reconcile_registry(infos)
                # reconcile firewall gateways against live state (reference:
                # dockerevents reconcile + FirewallEnable drift guard)
                live_fw = {i.name: i for i in running
                           if i.labels.get("dev.clawker.fw") == "on"
                           and i.name not in self._fw_disabled}
                for name, i in live_fw.items():
                    if name not in self.gateways.gateways:
                        self._attach_gateway(name, i.rundir)
                for name in list(self.gateways.gateways):
                    if name in live_fw:
                        continue
                    # TOCTOU guard: an admin fw_attach can land AFTER
                    # this tick's engine snapshot was taken — detaching
                    # on the stale set killed a fresh gateway the
                    # instant its agent sent its first request
                    # (observed 2/4000 at 8-way cold-start bursts).
                    gw = self.gateways.gateways.get(name)
                    if gw is not None and time.time() - gw.attached_at < 3.0:
                        continue
                    try:
                        cur = self.engine.inspect(name)
                        if (cur.state in ("running", "paused")
                                and cur.labels.get("dev.clawker.fw") == "on"
                                and name not in self._fw_disabled):
                            continue   # fresh state says keep it
                    except Exception:
                        pass
                    self.gateways.detach(name)
                # ssh/gpg agent bridges for every running sandbox
                running_names = {i.name for i in running}
                for i in running:
                    self.bridges.attach(i.name, i.rundir)
                for name in list(self.bridges._bridges):
                    if name not in running_names:
                        self.bridges.detach(name)
                self._maybe_drive_orphans(running)
                self.hbm.check(running)
                self._prune_rotated_events()
                if running:
                    self._last_agent_seen = time.time()
                elif (self.settings.control_plane.drain_to_zero and self.ready and
                      time.time() - self._last_agent_seen >
                      self.settings.control_plane.drain_grace_s):
                    log.info("drain_to_zero")
                    self.events.emit("cp_drain_to_zero")
                    self._stop.set()
            except Exception as e:
                reconcile_ok = False
                log.error("watcher_unavailable", err=str(e))
            # drain-to-zero must run even when a reconcile step throws
            # (observed: a deleted state dir made every tick raise, so a
            # test-spawned cpd lived forever); an erroring engine with no
            # confirmed agents counts as idle
            if running:
                self._last_agent_seen = time.time()
            elif (self.settings.control_plane.drain_to_zero and self.ready
                  and time.time() - self._last_agent_seen >
                  self.settings.control_plane.drain_grace_s):
                log.info("drain_to_zero", reconcile_ok=reconcile_ok)
                self.events.emit("cp_drain_to_zero")
                self._stop.set()
            self._stop.wait(1.0)

    _last_prune = 0.0

    def _prune_rotated_events(self) -> None:
        """settings monitoring.retention_hours: rotated event logs age
        out (the ISM retention analog, MONITORING-REFERENCE.md)."""
        now = time.time()
        if now - self._last_prune < 600:
            return
        self._last_prune = now
        keep_s = self.settings.monitoring.retention_hours * 3600
        rotated = events_path().with_suffix(".jsonl.1")
        try:
            if rotated.is_file() and now - rotated.stat().st_mtime > keep_s:
                rotated.unlink()
                self.events.emit("events_log_pruned", file=str(rotated))
        except OSError:
            pass

    # ------------------------------------------------------- orphan plans ---
    def _maybe_drive_orphans(self, running) -> None:
        """A gated (autostart=False) sandbox normally gets its Init/Boot
        plans from the starting client (CLI/fleet), which holds a ckd
        session while doing so. When that client is gone — clients==1
        (only our probe) and no CMD running past the grace — the CP
        drives the plans itself so `run -d` from a killed CLI still
        reaches AgentReady (reference: CP-side Executor,
        init_steps.go:67 / boot_steps.go:52)."""
        now = time.time()
        grace = getattr(self.settings.control_plane, "orphan_grace_s", 10)
        running_names = {i.name for i in running}
        for name in list(self._orphan_since):
            if name not in running_names:
                self._orphan_since.pop(name, None)
        for i in running:
            if i.name in self._driving or i.state != "running":
                continue
            try:
                spec = json.loads(
                    (Path(i.rundir) / "spec.json").read_text())
            except (OSError, ValueError):
                continue
            if spec.get("autostart"):
                continue
            try:
                with self.engine.client(i.name, timeout=2) as c:
                    h = c.hello()
            except Exception:
                continue
            if h.get("cmd_running") or int(h.get("clients", 99)) > 1:
                self._orphan_since.pop(i.name, None)
                continue
            first = self._orphan_since.setdefault(i.name, now)
            if now - first < grace:
                continue
            self._driving.add(i.name)
            threading.Thread(target=self._drive_plans, args=(i.name,),
                             daemon=True).start()

    def _drive_plans(self, name: str) -> None:
        try:
            from .plans import drive_boot
            with self.engine.client(name) as c:
                h = c.hello()
                if not h.get("cmd_running"):
                    drive_boot(c, h)
                    c.agent_ready()
                    self.events.emit("cp_plans_driven", sandbox=name)
                    log.info("orphan_plans_driven", sandbox=name)
        except Exception as e:
            log.error("orphan_plan_failed", sandbox=name, err=str(e))
        finally:
            self._driving.discard(name)
            self._orphan_since.pop(name, None)

    # -------------------------------------------------------------- policy --
    def _reload_policy(self) -> int:
        """Regenerate per-sandbox policy snapshots and nudge gateways
        (reference: reconcileStackClosure — config gen + stack reload +
        route sync)."""
        from ..firewall.policy import compile_policy, write_policy_snapshot
        n = 0
        try:
            pol = compile_policy(bypass=self.bypassed())
            for i in self.engine.list():
                if i.state == "running":
                    write_policy_snapshot(i.rundir, pol)
                    n += 1
            self.events.emit("policy_reloaded", sandboxes=n)
        except Exception as e:
            log.error("policy_reload_unavailable", err=str(e))
        return n

    def _attach_gateway(self, name: str, rundir) -> None:
        from pathlib import Path as _P
        from ..firewall.policy import compile_policy, write_policy_snapshot
        rundir = _P(rundir)
        write_policy_snapshot(rundir, compile_policy(bypass=self.bypassed()))
        self.gateways.attach(name, rundir)
        self.events.emit("firewall_enabled", sandbox=name)

    def bypassed(self) -> bool:
        return time.time() < self._bypass_until

    def _set_bypass(self, seconds: int) -> None:
        cap = self.settings.firewall.bypass_max_s
        seconds = max(0, min(seconds, cap))
        self._bypass_until = time.time() + seconds
        self.events.emit("firewall_bypass", seconds=seconds)
        if self._bypass_timer:
            self._bypass_timer.cancel()
        if seconds > 0:
            # dead-man restore (reference: Bypass 1h-capped timer)
            self._bypass_timer = threading.Timer(
                seconds, lambda: self.queue.submit("reload", self._reload_policy))
            self._bypass_timer.daemon = True
            self._bypass_timer.start()
        self.queue.submit("reload", self._reload_policy)

    # --------------------------------------------------------------- admin --
    def _handle_admin(self, req: dict) -> dict:
        op = req.get("op")
        if op == "ping":
            return {"ok": True, "ready": self.ready, "pid": os.getpid()}
        if op == "status":
            cur = self._registry_db.execute(
                "SELECT COUNT(*) FROM agents WHERE state='running'")
            return {"ok": True, "ready": self.ready,
                    "bypass": self.bypassed(),
                    "bypass_until": self._bypass_until,
                    "agents_running": cur.fetchone()[0]}
        if op == "agents":
            cur = self._registry_db.execute(
                "SELECT sandbox,project,agent,first_seen,last_seen,state FROM agents")
            cols = ["sandbox", "project", "agent", "first_seen", "last_seen", "state"]
            return {"ok": True,
                    "agents": [dict(zip(cols, r)) for r in cur.fetchall()]}
        if op == "reload_policy":
            n = self._reload_policy()
            return {"ok": True, "sandboxes": n}
        if op == "bypass":
            self._set_bypass(int(req.get("seconds", 0)))
            return {"ok": True, "until": self._bypass_until}
        if op == "fw_add_rules":
            # single-writer rule mutation through the action queue
            # (reference: FirewallAddRules -> ActionQueue RuleMutate)
            from ..config.schema import EgressRule
            from ..firewall import EgressRulesStore
            from ..storage import materialize
            rules = [materialize(EgressRule, r) for r in req.get("rules", [])]

            def mutate():
                changed = EgressRulesStore().add(rules)
                if changed:
                    self._reload_policy()
                return changed
            changed = self.queue.submit_sync("rule_mutate", mutate)
            return {"ok": True, "changed": bool(changed)}
        if op == "fw_attach":
            self._fw_disabled.discard(req["sandbox"])
            self._attach_gateway(req["sandbox"], req["rundir"])
            return {"ok": True}
        if op == "fw_bootstrap":
            # rule push + gateway attach in ONE round trip (the cold-start
            # path calls this per sandbox; separate ops cost 2-3 RTTs)
            from ..config.schema import EgressRule
            from ..firewall import EgressRulesStore
            from ..storage import materialize
            rules = [materialize(EgressRule, r) for r in req.get("rules", [])]
            changed = False
            if rules:
                def mutate():
                    ch = EgressRulesStore().add(rules)
                    if ch:
                        self._reload_policy()
                    return ch
                changed = self.queue.submit_sync("rule_mutate", mutate)
            self._fw_disabled.discard(req["sandbox"])
            self._attach_gateway(req["sandbox"], req["rundir"])
            return {"ok": True, "changed": bool(changed)}
        if op == "fw_detach":
            if req.get("sticky"):
                self._fw_disabled.add(req["sandbox"])
            self.gateways.detach(req["sandbox"])
            return {"ok": True}
        if op == "fw_status":
            return {"ok": True,
                    "gateways": sorted(self.gateways.gateways.keys()),
                    "disabled": sorted(self._fw_disabled),
                    "bypass": self.bypassed(),
                    # rate-limited event drops per sandbox (reference:
                    # events_drops counter map next to the ringbuf)
                    "event_drops": {
                        n: g.bucket.dropped
                        for n, g in self.gateways.gateways.items()
                        if g.bucket.dropped},
                    "active_tunnels": len(
                        getattr(self.gateways, "_tunnels", []))}
        if op == "events":
            return {"ok": True, "events": self.events.tail(int(req.get("n", 100)))}
        if op == "shutdown":
            self._stop.set()
            return {"ok": True}
        return {"ok": False, "error": f"unknown op: {op}"}

    def _admin_loop(self, listener: socket.socket) -> None:
        listener.settimeout(0.5)
        while not self._stop.is_set():
            try:
                conn, _ = listener.accept()
            except socket.timeout:
                continue
            except OSError:
                break
            threading.Thread(target=self._serve_conn, args=(conn,),
                             daemon=True).start()

    def _follow_events(self, conn: socket.socket, req: dict) -> None:
        """Stream events as frames until the client goes away (reference:
        pub/sub subscriber semantics: bounded buffer, drop-oldest)."""
        sub = self.topic.subscribe()
        try:
            wire.send_frame(conn, {"ok": True, "stream": True})
            while not self._stop.is_set():
                ev = sub.get(timeout=1.0)
                if ev is None:
                    continue
                wire.send_frame(conn, {"event": ev, "dropped": sub.dropped})
        except OSError:
            pass          # client disconnected
        finally:
            sub.close()

    def _serve_conn(self, conn: socket.socket) -> None:
        try:
            conn.settimeout(30)
            while True:
                req = wire.recv_frame(conn)
                if req is None:
                    return
                if req.get("op") == "events_follow":
                    conn.settimeout(None)
                    self._follow_events(conn, req)
                    return
                try:
                    resp = self._handle_admin(req)
                except Exception as e:   # degrade, never crash
                    log.error("admin_op_failed", op=req.get("op"), err=str(e))
                    resp = {"ok": False, "error": str(e)}
                wire.send_frame(conn, resp)
        except OSError:
            pass
        finally:
            conn.close()

    # ----------------------------------------------------------------- run --
    def run(self) -> int:
        rd = consts.runtime_dir()
        rd.mkdir(parents=True, exist_ok=True)
        sock_path = admin_sock_path()
        # never unlink a LIVE sibling's socket (a spawn race would
        # otherwise leave two daemons, the newer stealing the address):
        # if someone answers ping, we are redundant — exit cleanly
        try:
            probe = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            probe.settimeout(1.0)
            probe.connect(str(sock_path))
            from ..engine import wire as _w
            _w.send_frame(probe, {"op": "ping"})
            if (_w.recv_frame(probe) or {}).get("ok"):
                probe.close()
                log.info("cpd_already_running")
                return 0
            probe.close()
        except OSError:
            pass
        listener = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        try:
            sock_path.unlink(missing_ok=True)
            listener.bind(str(sock_path))
            os.chmod(sock_path, 0o600)
            # unix SOCK_STREAM connect() fails with EAGAIN the moment the
            # accept queue is full (no TCP-style SYN retry), so size the
            # backlog for a whole fleet cold-starting at once: 32 agents ×
            # several CP round-trips each showed EAGAIN bursts at 16
            listener.listen(512)
        except OSError as e:
            log.error("admin_listen_failed", err=str(e))
            return 1
        pid_path().write_text(str(os.getpid()))
        self.events.emit("cp_starting", pid=os.getpid())

        threads = [
            threading.Thread(target=self._admin_loop, args=(listener,), daemon=True),
            threading.Thread(target=self._watch_loop, daemon=True),
        ]
        for t in threads:
            t.start()
        self.queue.submit("reload", self._reload_policy)
        self.ready = True
        self.events.emit("cp_ready")

        def on_sig(*_a):
            self._stop.set()
        signal.signal(signal.SIGTERM, on_sig)
        signal.signal(signal.SIGINT, on_sig)

        while not self._stop.is_set():
            self._stop.wait(0.5)

        # drain sequence (reference: runDrainSequence ordering — queue
        # close -> stop serving -> cancel bypass timers -> stack stop)
        self.events.emit("cp_draining")
        self.queue.close()
        if self._bypass_timer:
            self._bypass_timer.cancel()
        self.gateways.detach_all()
        self.bridges.detach_all()
        listener.close()
        sock_path.unlink(missing_ok=True)
        pid_path().unlink(missing_ok=True)
        self.events.emit("cp_stopped")
        return 0


def main() -> int:
    from ..config.config import load_settings
    try:
        ls = load_settings().get().logging
        logger_setup(consts.log_dir() / "cpd.log", level=ls.level,
                     max_size_mb=ls.max_size_mb, max_backups=ls.max_backups)
    except Exception:
        logger_setup(consts.log_dir() / "cpd.log")
    return CPDaemon().run()


if __name__ == "__main__":
    raise SystemExit(main())
