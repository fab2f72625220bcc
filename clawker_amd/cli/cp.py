"""Control-plane verbs (reference: internal/cmd/controlplane up/down/
status/agents)."""
from __future__ import annotations

import json

import click

from .root import Ctx, cli, pass_factory


@cli.group("controlplane")
def cp_group():
    """Control-plane daemon (cpd)."""


@cp_group.command("up")
@pass_factory
def cp_up(ctx: Ctx):
    """Start the control-plane daemon (idempotent)."""
    cp = ctx.factory.controlplane()
    cp.ensure_running()
    ctx.factory.io.success("control plane ready")


@cp_group.command("down")
@pass_factory
def cp_down(ctx: Ctx):
    """Stop the control-plane daemon (sandboxes keep running; egress fails closed)."""
    cp = ctx.factory.controlplane()
    cp.auto_start = False
    if cp.stop():
        ctx.factory.io.success("control plane stopped")
    else:
        ctx.factory.io.eprint("control plane not running")


@cp_group.command("status")
@pass_factory
def cp_status(ctx: Ctx):
    """Daemon liveness, bypass state and agent count."""
    cp = ctx.factory.controlplane()
    cp.auto_start = False
    if not cp.running():
        ctx.factory.io.print(json.dumps({"running": False}))
        return
    ctx.factory.io.print(json.dumps({"running": True, **cp.status()}, indent=1))


@cp_group.command("agents")
@click.option("-a", "--all", "show_all", is_flag=True,
              help="include rows for removed sandboxes")
@pass_factory
def cp_agents(ctx: Ctx, show_all):
    """Agent registry (cpd's sqlite; rows close out as 'removed' when
    their sandbox leaves the engine)."""
    cp = ctx.factory.controlplane()
    rows = cp.agents()
    if not show_all:
        rows = [r for r in rows if r.get("state") != "removed"]
    ctx.factory.io.print(json.dumps(rows, indent=1))


@cp_group.command("events")
@click.option("-n", type=int, default=50, show_default=True)
@click.option("-f", "--follow", is_flag=True)
@pass_factory
def cp_events(ctx: Ctx, n, follow):
    """Tail (or follow) the control-plane event stream."""
    cp = ctx.factory.controlplane()
    for ev in cp.events(n):
        ctx.factory.io.print(json.dumps(ev))
    if not follow:
        return
    # live push stream via cpd pub/sub (reference: Topic subscriber)
    for ev in cp.follow_events():
        ctx.factory.io.print(json.dumps(ev))


@cp_group.command("serve", hidden=True)
def cp_serve():
    """Run the control-plane daemon in the foreground (hidden; normally
    spawned on demand — reference: hidden hostproxy/bridge daemons)."""
    import sys
    from ..controlplane.daemon import main as cpd_main
    sys.exit(cpd_main())


# docker-style top-level alias: `clawker events [-f]`
cli.add_command(cp_events, "events")
