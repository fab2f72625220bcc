"""Volume + auth verbs (reference: internal/cmd/volume, `clawker auth
rotate`)."""
from __future__ import annotations

import json
import time

import click

from ..cmdutil import format_age
from .root import Ctx, cli, pass_factory


@cli.group("volume")
def volume_group():
    """Named volumes (agent config/history state; survive sandbox rm)."""


@volume_group.command("ls")
@click.option("--format", "fmt", default="")
@pass_factory
def volume_ls(ctx: Ctx, fmt):
    f = ctx.factory
    vols = f.engine().db.list_volumes()
    if fmt == "json":
        f.io.print(json.dumps(vols, indent=1))
        return
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for c in ("NAME", "PATH", "AGE"):
        t.add_column(c)
    now = time.time()
    for v in vols:
        t.add_row(v["name"], v["path"], format_age(now - v["created"]))
    f.io.print(t)


@volume_group.command("inspect")
@click.argument("names", nargs=-1, required=True)
@pass_factory
def volume_inspect(ctx: Ctx, names):
    """Volume details incl. disk usage (docker volume inspect analog)."""
    import subprocess
    from pathlib import Path as _P
    f = ctx.factory
    vols = {v["name"]: v for v in f.engine().db.list_volumes()}
    out = []
    for n in names:
        full = n if n.startswith("clawker.") else f"clawker.user.{n}"
        v = vols.get(full) or vols.get(n)
        if v is None:
            from ..errors import NotFoundError
            raise NotFoundError(f"volume not found: {n}")
        du = subprocess.run(["du", "-sb", v["path"]], capture_output=True,
                            text=True)
        size = int(du.stdout.split()[0]) if du.returncode == 0 else None
        out.append({**v, "size_bytes": size,
                    "exists": _P(v["path"]).is_dir()})
    f.io.print(json.dumps(out, indent=1))


@volume_group.command("create")
@click.argument("name")
@pass_factory
def volume_create(ctx: Ctx, name):
    """Create a named volume (mount with -v NAME:/dst)."""
    f = ctx.factory
    full = name if name.startswith("clawker.") else f"clawker.user.{name}"
    path, fresh = f.engine().ensure_volume(full, {})
    f.io.print(full if fresh else f"{full} (exists)")


@volume_group.command("rm")
@click.argument("names", nargs=-1, required=True)
@pass_factory
def volume_rm(ctx: Ctx, names):
    f = ctx.factory
    for n in names:
        f.engine().remove_volume(n)
        f.io.print(n)


@volume_group.command("prune")
@pass_factory
def volume_prune(ctx: Ctx):
    """Remove volumes not referenced by any sandbox."""
    f = ctx.factory
    eng = f.engine()
    live = {i.name for i in eng.list()}
    removed = 0
    for v in eng.db.list_volumes():
        owner = v["name"].rsplit("-", 1)[0]
        if v["name"].startswith("clawker.") and owner not in live \
                and not v["name"].endswith(".share"):
            eng.remove_volume(v["name"])
            f.io.print(v["name"])
            removed += 1
    f.io.eprint(f"removed {removed} volume(s)")


@cli.group("auth")
def auth_group():
    """Agent identity trust material."""


@auth_group.command("rotate")
@pass_factory
def auth_rotate(ctx: Ctx):
    """Rotate the root key; existing agent tokens become invalid
    (restart sandboxes to re-mint). Reference: `clawker auth rotate`."""
    from .. import auth
    auth.rotate_auth_material()
    ctx.factory.io.success("auth material rotated (restart agents to re-mint tokens)")
