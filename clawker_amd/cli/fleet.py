"""Fleet verbs: the autonomous-loop driver (BASELINE configs 3-5)."""
from __future__ import annotations

import json
import time

import click

from ..fleet import Fleet, FleetOptions
from .root import Ctx, cli, pass_factory


@cli.group("fleet")
def fleet_group():
    """N parallel agent loops, one pinned MI355X each."""


@fleet_group.command("up")
@click.option("-n", "--count", type=int, default=2, show_default=True)
@click.option("--branch-prefix", default="agent", show_default=True)
@click.option("--gpus-per-agent", type=int, default=None)
@click.option("--image", default="")
@click.option("--base", default="", help="worktree start point")
@click.option("--no-worktrees", is_flag=True, help="snapshot workspaces instead")
@click.option("--firewall/--no-firewall", "firewall", default=None)
@click.option("--prompt", "prompt_file", default="",
              help="one-shot prompt file handed to each agent (harness prompt_cmd)")
@click.option("--watch", is_flag=True, help="open the live dashboard after start")
@click.argument("cmd", nargs=-1, type=click.UNPROCESSED)
@pass_factory
def fleet_up(ctx: Ctx, count, branch_prefix, gpus_per_agent, image, base,
             no_worktrees, firewall, prompt_file, watch, cmd):
    """Start N agent sandboxes over worktrees with 1:1 GPU pinning."""
    f = ctx.factory
    cfg = f.config(require_project=True)
    fleet = Fleet(cfg, f.orchestrator())
    if prompt_file:
        from .prompt import resolve_prompt
        prompt_file = str(resolve_prompt(prompt_file))
    members = fleet.up(FleetOptions(
        count=count, branch_prefix=branch_prefix, gpus_per_agent=gpus_per_agent,
        cmd=list(cmd), image=image, firewall=firewall, base=base,
        prompt_file=prompt_file,
        use_worktrees=False if no_worktrees else None))
    for m in members:
        gpus = ",".join(map(str, m.gpus)) or "-"
        f.io.print(f"{m.sandbox}  gpu={gpus}  branch={m.branch or '-'}")
    if watch:
        _watch(f)


@fleet_group.command("run")
@click.option("-n", "--count", type=int, default=2, show_default=True)
@click.option("--branch-prefix", default="agent", show_default=True)
@click.option("--gpus-per-agent", type=int, default=None)
@click.option("--prompt", "prompt_file", default="")
@click.option("--no-worktrees", is_flag=True)
@click.option("--firewall/--no-firewall", "firewall", default=None)
@click.option("--timeout", type=float, default=3600.0, show_default=True)
@click.option("--keep", is_flag=True, help="keep sandboxes after completion")
@click.argument("cmd", nargs=-1, type=click.UNPROCESSED)
@pass_factory
def fleet_run(ctx: Ctx, count, branch_prefix, gpus_per_agent, prompt_file,
              no_worktrees, firewall, timeout, keep, cmd):
    """up + wait + report (+ down): the one-shot autonomous-loop driver."""
    f = ctx.factory
    cfg = f.config(require_project=True)
    fleet = Fleet(cfg, f.orchestrator())
    if prompt_file:
        from .prompt import resolve_prompt
        prompt_file = str(resolve_prompt(prompt_file))
    members = fleet.up(FleetOptions(
        count=count, branch_prefix=branch_prefix, gpus_per_agent=gpus_per_agent,
        cmd=list(cmd), firewall=firewall, prompt_file=prompt_file,
        use_worktrees=False if no_worktrees else None))
    fleet.wait(members, timeout_s=timeout)
    worst = 0
    for m in members:
        mark = "[green]✓[/green]" if m.exit_code == 0 else "[red]✗[/red]"
        f.io.print(f" {mark} {m.sandbox}  exit={m.exit_code}  "
                   f"branch={m.branch or '-'}")
        worst = max(worst, m.exit_code or 0)
    if not keep:
        fleet.down(branch_prefix)
    if worst:
        from ..errors import ExitError
        raise ExitError(worst)


@fleet_group.command("status")
@click.option("--branch-prefix", default="agent", show_default=True)
@click.option("--format", "fmt", default="")
@pass_factory
def fleet_status(ctx: Ctx, branch_prefix, fmt):
    f = ctx.factory
    fleet = Fleet(f.config(require_project=True), f.orchestrator())
    members = fleet.status(branch_prefix)
    if fmt == "json":
        f.io.print(json.dumps([m.__dict__ for m in members], indent=1))
        return
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for c in ("SANDBOX", "STATE", "GPUS", "EXIT"):
        t.add_column(c)
    for m in members:
        t.add_row(m.sandbox, m.state, ",".join(map(str, m.gpus)) or "-",
                  str(m.exit_code) if m.exit_code is not None else "-")
    f.io.print(t)


@fleet_group.command("wait")
@click.option("--branch-prefix", default="agent", show_default=True)
@click.option("--timeout", type=float, default=3600.0)
@pass_factory
def fleet_wait(ctx: Ctx, branch_prefix, timeout):
    f = ctx.factory
    fleet = Fleet(f.config(require_project=True), f.orchestrator())
    members = fleet.wait(fleet.status(branch_prefix), timeout_s=timeout)
    worst = 0
    for m in members:
        f.io.print(f"{m.sandbox} exit={m.exit_code}")
        worst = max(worst, m.exit_code or 0)
    if worst:
        from ..errors import ExitError
        raise ExitError(worst)


@fleet_group.command("down")
@click.option("--branch-prefix", default="agent", show_default=True)
@click.option("--keep", is_flag=True, help="stop but do not remove")
@pass_factory
def fleet_down(ctx: Ctx, branch_prefix, keep):
    f = ctx.factory
    fleet = Fleet(f.config(require_project=True), f.orchestrator())
    n = fleet.down(branch_prefix, remove=not keep)
    f.io.success(f"{'stopped' if keep else 'removed'} {n} fleet member(s)")


def _watch(f):
    from rich.live import Live
    from ..monitor.stats import collect_stats
    from ..tui.dashboard import render_dashboard
    with Live(console=f.io.console, refresh_per_second=4, screen=True) as live:
        try:
            while True:
                live.update(render_dashboard(collect_stats(f.engine())))
                time.sleep(1.0)
        except KeyboardInterrupt:
            pass
