"""Full-screen dashboard renderer: agents x GPUs panes (reference:
internal/tui dashboard.go DashboardRenderer + the bypass countdown and
stats precedents)."""
from __future__ import annotations

import time

from rich.console import Group
from rich.panel import Panel
from rich.progress_bar import ProgressBar
from rich.table import Table

from ..monitor.stats import StatsSnapshot


def render_dashboard(snap: StatsSnapshot):
    panes = []

    gt = Table(box=None, expand=True, pad_edge=False)
    for c in ("GPU", "BUSY", "", "VRAM", "POWER", "TEMP(J)", "OWNER"):
        gt.add_column(c)
    for g in snap.gpus:
        bar = ProgressBar(total=100, completed=g.busy_pct, width=20)
        vram_pct = 100 * g.vram_used / g.vram_total if g.vram_total else 0
        gt.add_row(
            f"[bold]{g.index}[/bold]", f"{g.busy_pct:3.0f}%", bar,
            f"{g.vram_used / 2**30:6.1f}G ({vram_pct:2.0f}%)",
            f"{g.power_w:5.0f}W", f"{g.temp_junction_c:3.0f}C",
            snap.allocations.get(g.index, "[dim]free[/dim]"))
    panes.append(Panel(gt, title="MI355X GPUs", border_style="cyan"))

    st = Table(box=None, expand=True, pad_edge=False)
    for c in ("SANDBOX", "STATE", "MEM", "GPUS", "CONSOLE"):
        st.add_column(c)
    for s in snap.sandboxes:
        mem = s.get("mem_bytes")
        state_style = {"running": "green", "exited": "dim"}.get(s["state"], "yellow")
        st.add_row(s["name"], f"[{state_style}]{s['state']}[/{state_style}]",
                   f"{mem / 2**20:.0f}M" if mem else "-",
                   ",".join(map(str, s["gpus"])) or "-",
                   f"[dim]{s.get('tail', '')}[/dim]")
    panes.append(Panel(st, title="agent sandboxes", border_style="magenta"))

    if snap.events:
        et = Table(box=None, expand=True, pad_edge=False)
        for c in ("TIME", "SANDBOX", "ACTION", "DST"):
            et.add_column(c)
        for e in snap.events:
            act = e.get("action", "")
            style = "green" if act in ("allow", "resolve") else "red"
            dst = e.get("dst", "")
            if e.get("path"):
                dst += e["path"]
            et.add_row(time.strftime("%H:%M:%S", time.localtime(e.get("ts", 0))),
                       str(e.get("sandbox", ""))[-24:],
                       f"[{style}]{act}[/{style}]", dst[:48])
        panes.append(Panel(et, title="egress decisions", border_style="yellow"))
    panes.append(f"[dim]{time.strftime('%H:%M:%S', time.localtime(snap.ts))} — "
                 f"ctrl-c to exit[/dim]")
    return Group(*panes)
