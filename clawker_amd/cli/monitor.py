"""Monitoring verbs: live GPU + sandbox telemetry.

Reference: internal/cmd/monitor (compose observability stack) and
internal/cmd/container/stats (streamStats repaint loop). MI355X-first:
`clawker stats` streams a live table of per-sandbox cgroup stats and
per-GPU rocm telemetry from the zero-spawn native sampler; `clawker gpus`
shows the inventory/allocation map; `clawker monitor up` starts the
Prometheus exporter daemon."""
from __future__ import annotations

import json
import shutil
import time

import click

from .. import consts
from ..cmdutil import format_age
from ..errors import ClawkerError
from .root import Ctx, cli, pass_factory


@cli.command("gpus")
@click.option("--format", "fmt", default="")
@pass_factory
def gpus_cmd(ctx: Ctx, fmt):
    """GPU inventory and allocations."""
    f = ctx.factory
    orch = f.orchestrator()
    inv = orch.allocator.inventory
    allocs = orch.allocator.allocations()
    rows = []
    samples = {}
    try:
        from ..monitor import RocmSampler
        for s in RocmSampler(inv).sample():
            samples[s.index] = s
    except Exception:
        pass
    for d in inv.devices:
        s = samples.get(d.index)
        rows.append({
            "index": d.index, "render": d.render_path, "pci": d.pci_bus,
            "vram_gb": round(d.vram_total / 2**30, 1),
            "owner": allocs.get(d.index, ""),
            "busy_pct": s.busy_pct if s else None,
            "vram_used_gb": round(s.vram_used / 2**30, 1) if s else None,
            "power_w": s.power_w if s else None,
            "temp_c": s.temp_junction_c or s.temp_edge_c if s else None,
            "xgmi_peers": d.xgmi_peers,
        })
    if fmt == "json":
        f.io.print(json.dumps(rows, indent=1))
        return
    from ..tui.components import plain_table
    f.io.print(plain_table(
        ("GPU", "RENDER", "VRAM", "BUSY", "USED", "POWER", "TEMP", "OWNER"),
        [(r["index"], r["render"], f"{r['vram_gb']}G",
          f"{r['busy_pct']:.0f}%" if r["busy_pct"] is not None else None,
          f"{r['vram_used_gb']}G" if r["vram_used_gb"] is not None else None,
          f"{r['power_w']:.0f}W" if r["power_w"] is not None else None,
          f"{r['temp_c']:.0f}C" if r["temp_c"] else None,
          r["owner"]) for r in rows]))


@cli.command("stats")
@click.option("--no-stream", is_flag=True, help="print once and exit")
@click.option("--format", "fmt", default="", help="json for machine output")
@click.option("--interval", type=float, default=None,
              help="refresh seconds (default: settings "
                   "monitoring.sample_interval_ms)")
@pass_factory
def stats_cmd(ctx: Ctx, no_stream, fmt, interval):
    """Live per-sandbox + per-GPU stats (replaces docker stats)."""
    if interval is None:
        interval = max(0.1, ctx.factory.config().settings
                       .monitoring.sample_interval_ms / 1000.0)
    f = ctx.factory
    from ..monitor.stats import collect_stats, render_stats
    if fmt == "json":
        import dataclasses
        snap = collect_stats(f.engine())
        f.io.print(json.dumps({
            "gpus": [dataclasses.asdict(g) for g in snap.gpus],
            "allocations": snap.allocations,
            "sandboxes": snap.sandboxes,
        }, indent=1, default=str))
        return
    if no_stream or not f.io.is_stdout_tty():
        snap = collect_stats(f.engine())
        f.io.print(render_stats(snap))
        return
    from rich.live import Live
    with Live(console=f.io.console, refresh_per_second=4) as live:
        while True:
            snap = collect_stats(f.engine())
            live.update(render_stats(snap))
            time.sleep(interval)


@cli.group("monitor")
def monitor_group():
    """Telemetry exporter + dashboards."""


@monitor_group.command("up")
@pass_factory
def monitor_up(ctx: Ctx):
    """Start the Prometheus metrics exporter daemon."""
    from ..monitor.exporter import ensure_running
    port = ctx.factory.config().settings.monitoring.prometheus_port
    ensure_running(port)
    ctx.factory.io.success(f"metrics exporter on :{port}/metrics")


@monitor_group.command("down")
@pass_factory
def monitor_down(ctx: Ctx):
    from ..monitor.exporter import stop_running
    if stop_running():
        ctx.factory.io.success("metrics exporter stopped")
    else:
        ctx.factory.io.eprint("metrics exporter not running")


@monitor_group.command("status")
@pass_factory
def monitor_status(ctx: Ctx):
    from ..monitor.exporter import exporter_running
    port = ctx.factory.config().settings.monitoring.prometheus_port
    ctx.factory.io.print(json.dumps(
        {"running": exporter_running(), "port": port}))


@cli.command("dashboard")
@click.option("--interval", type=float, default=None,
              help="refresh seconds (default: settings "
                   "monitoring.sample_interval_ms)")
@pass_factory
def dashboard_cmd(ctx: Ctx, interval):
    """Full-screen live dashboard: agents x GPUs (reference: tui
    RunDashboard precedent)."""
    if interval is None:
        interval = max(0.1, ctx.factory.config().settings
                       .monitoring.sample_interval_ms / 1000.0)
    f = ctx.factory
    from ..monitor.stats import collect_stats
    from ..tui.dashboard import render_dashboard
    from rich.live import Live
    with Live(console=f.io.console, refresh_per_second=4, screen=True) as live:
        try:
            while True:
                snap = collect_stats(f.engine())
                live.update(render_dashboard(snap))
                time.sleep(interval)
        except KeyboardInterrupt:
            pass


@monitor_group.command("extensions")
@click.option("--install", "install_file", default=None,
              type=click.Path(exists=True, dir_okay=False),
              help="drop a .prom collector file into the metrics.d lane")
@click.option("--remove", "remove_name", default="",
              help="remove an installed collector by name")
@pass_factory
def monitor_extensions(ctx: Ctx, install_file, remove_name):
    """Textfile-collector extensions (reference: monitoring units plugin
    model — here: .prom files in <state>/metrics.d/ merged into /metrics)."""
    import time as _t
    from pathlib import Path as _P
    f = ctx.factory
    metrics_d = consts.state_dir() / "metrics.d"
    if install_file:
        src = _P(install_file)
        if src.suffix != ".prom":
            raise ClawkerError("collector files must end in .prom")
        metrics_d.mkdir(parents=True, exist_ok=True)
        shutil.copy2(src, metrics_d / src.name)
        f.io.success(f"installed {src.name} -> {metrics_d}")
        return
    if remove_name:
        p = metrics_d / (remove_name if remove_name.endswith(".prom")
                         else remove_name + ".prom")
        if not p.is_file():
            raise ClawkerError(f"no such collector: {p.name}")
        p.unlink()
        f.io.success(f"removed {p.name}")
        return
    files = sorted(metrics_d.glob("*.prom")) if metrics_d.is_dir() else []
    if not files:
        f.io.eprint(f"no extensions ({metrics_d}/*.prom)")
        return
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for c in ("NAME", "METRICS", "AGE"):
        t.add_column(c)
    for p in files:
        try:
            text = p.read_text()
            n = sum(1 for l in text.splitlines()
                    if l.strip() and not l.startswith("#"))
            age = format_age(_t.time() - p.stat().st_mtime)
        except OSError:
            n, age = 0, "?"
        t.add_row(p.name, str(n), age)
    f.io.print(t)


@monitor_group.command("serve", hidden=True)
@click.option("--port", type=int, default=19090)
def monitor_serve(port):
    """Run the metrics exporter in the foreground (hidden)."""
    import sys
    from ..monitor.exporter import serve
    sys.exit(serve(port))


@cli.command("hostproxy", hidden=True)
def hostproxy_serve():
    """Run the hostproxy daemon in the foreground (hidden; reference:
    `clawker host-proxy serve`)."""
    import sys
    from ..hostproxy.daemon import main as hp_main
    sys.exit(hp_main())
