"""Prometheus metrics exporter daemon: /metrics over HTTP on loopback.

Reference: the compose monitoring stack's Prometheus scrape of the OTel
collector (internal/monitor). Single-node redesign: one small exporter
serving amdgpu + sandbox-cgroup gauges straight from the zero-spawn
samplers — no collector chain.
"""
from __future__ import annotations

import json
import os
import signal
import subprocess
import sys
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from pathlib import Path

from .. import consts
from ..logger import get as get_logger

log = get_logger("exporter")


def pid_path() -> Path:
    return consts.runtime_dir() / "exporter.pid"


class _EventCounters:
    """Incremental tail of cp-events.jsonl -> monotonic counters (the
    netlogger/OTLP egress-decision lane, re-landed as Prometheus metrics:
    this node's observability sink is the scrape endpoint, not a
    collector). Rotation-aware: the 10MB rollover resets the offset."""

    def __init__(self):
        self.offset = 0
        self.ino = None
        self.events: dict[str, int] = {}
        self.egress: dict[tuple, int] = {}

    def update(self) -> None:
        from ..controlplane.daemon import events_path
        path = events_path()
        try:
            st = path.stat()
        except OSError:
            return
        if st.st_ino != self.ino or st.st_size < self.offset:
            self.ino, self.offset = st.st_ino, 0    # new/rotated file
        try:
            with open(path) as f:
                f.seek(self.offset)
                chunk = f.read()
                self.offset = f.tell()
        except OSError:
            return
        for line in chunk.splitlines():
            try:
                rec = json.loads(line)
            except ValueError:
                continue
            ev = rec.get("event", "unknown")
            self.events[ev] = self.events.get(ev, 0) + 1
            if ev == "egress_decision":
                key = (rec.get("sandbox", ""), rec.get("action", ""),
                       rec.get("proto", ""))
                self.egress[key] = self.egress.get(key, 0) + 1

    def lines(self) -> list[str]:
        out = ["# TYPE clawker_cp_events_total counter"]
        for ev, n in sorted(self.events.items()):
            out.append(f'clawker_cp_events_total{{event="{ev}"}} {n}')
        if self.egress:
            out.append("# TYPE clawker_egress_decisions_total counter")
            for (sb, action, proto), n in sorted(self.egress.items()):
                out.append(
                    f'clawker_egress_decisions_total{{sandbox="{sb}",'
                    f'action="{action}",proto="{proto}"}} {n}')
        return out


_event_counters = _EventCounters()


def _metrics_text() -> str:
    from ..engine import Engine
    from .stats import collect_stats
    eng = Engine()
    try:
        snap = collect_stats(eng)
    finally:
        eng.close()
    lines = [
        "# HELP clawker_gpu_busy_percent amdgpu busy percent",
        "# TYPE clawker_gpu_busy_percent gauge",
    ]
    for g in snap.gpus:
        lbl = f'gpu="{g.index}"'
        owner = snap.allocations.get(g.index, "")
        if owner:
            lbl += f',sandbox="{owner}"'
        lines += [
            f"clawker_gpu_busy_percent{{{lbl}}} {g.busy_pct}",
            f"clawker_gpu_vram_used_bytes{{{lbl}}} {g.vram_used}",
            f"clawker_gpu_vram_total_bytes{{{lbl}}} {g.vram_total}",
            f"clawker_gpu_power_watts{{{lbl}}} {g.power_w}",
            f"clawker_gpu_temp_junction_celsius{{{lbl}}} {g.temp_junction_c}",
            f"clawker_gpu_sclk_mhz{{{lbl}}} {g.sclk_mhz}",
        ]
    running = 0
    for s in snap.sandboxes:
        lbl = f'sandbox="{s["name"]}",project="{s["project"]}",agent="{s["agent"]}"'
        state_v = 1 if s["state"] == "running" else 0
        running += state_v
        lines.append(f"clawker_sandbox_running{{{lbl}}} {state_v}")
        if s.get("mem_bytes"):
            lines.append(f"clawker_sandbox_memory_bytes{{{lbl}}} {s['mem_bytes']}")
        if s.get("pids"):
            lines.append(f"clawker_sandbox_pids{{{lbl}}} {s['pids']}")
        if s.get("vram_bytes") is not None:
            # per-sandbox drm-fdinfo VRAM attribution (monitor/hbm.py)
            lines.append(
                f"clawker_sandbox_vram_bytes{{{lbl}}} {s['vram_bytes']}")
    lines.append(f"clawker_sandboxes_running {running}")
    _event_counters.update()
    lines.extend(_event_counters.lines())
    # plugin lane: merge third-party textfile metrics (the reference's
    # monitoring-units collector routing, reduced to the node-exporter
    # textfile pattern: drop .prom files into <state>/metrics.d/)
    metrics_d = consts.state_dir() / "metrics.d"
    if metrics_d.is_dir():
        for p in sorted(metrics_d.glob("*.prom")):
            try:
                lines.append(f"# collector: {p.name}")
                lines.append(p.read_text().strip())
            except OSError:
                continue
    return "\n".join(lines) + "\n"


class _Handler(BaseHTTPRequestHandler):
    def do_GET(self):  # noqa: N802
        if self.path != "/metrics":
            self.send_response(404)
            self.end_headers()
            return
        try:
            body = _metrics_text().encode()
        except Exception as e:
            self.send_response(500)
            self.end_headers()
            self.wfile.write(str(e).encode())
            return
        self.send_response(200)
        self.send_header("Content-Type", "text/plain; version=0.0.4")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def log_message(self, *a):   # quiet
        pass


def serve(port: int) -> int:
    srv = ThreadingHTTPServer(("127.0.0.1", port), _Handler)
    pid_path().parent.mkdir(parents=True, exist_ok=True)
    pid_path().write_text(str(os.getpid()))
    log.info("exporter_listening", port=port)
    stop = threading.Event()
    signal.signal(signal.SIGTERM, lambda *a: stop.set())
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    while not stop.is_set():
        stop.wait(0.5)
    srv.shutdown()
    pid_path().unlink(missing_ok=True)
    return 0


def exporter_running() -> bool:
    try:
        pid = int(pid_path().read_text())
        os.kill(pid, 0)
        return True
    except (OSError, ValueError):
        return False


def ensure_running(port: int) -> None:
    if exporter_running():
        return
    consts.log_dir().mkdir(parents=True, exist_ok=True)
    logf = open(consts.log_dir() / "exporter.out", "ab")
    subprocess.Popen(
        [sys.executable, "-m", "clawker_amd.monitor.exporter", str(port)],
        stdin=subprocess.DEVNULL, stdout=logf, stderr=logf,
        start_new_session=True,
        cwd=str(Path(__file__).resolve().parents[2]))
    logf.close()
    deadline = time.monotonic() + 5
    while time.monotonic() < deadline:
        if exporter_running():
            return
        time.sleep(0.05)
    raise RuntimeError("exporter failed to start")


def stop_running() -> bool:
    if not exporter_running():
        return False
    try:
        os.kill(int(pid_path().read_text()), signal.SIGTERM)
        return True
    except (OSError, ValueError):
        return False


if __name__ == "__main__":
    sys.exit(serve(int(sys.argv[1]) if len(sys.argv) > 1 else 19090))
