"""System-wide verbs: disk usage + prune (docker `system df/prune`
ergonomics over the engine's stores)."""
from __future__ import annotations

import json
import shutil
from pathlib import Path

import click

from .. import consts
from .root import Ctx, cli, pass_factory


def _du(path: Path) -> int:
    total = 0
    try:
        for p in path.rglob("*"):
            try:
                if p.is_file() and not p.is_symlink():
                    total += p.stat().st_size
            except OSError:
                continue
    except OSError:
        pass
    return total


@cli.group("system")
def system_group():
    """System-wide information and cleanup."""


@system_group.command("df")
@click.option("--format", "fmt", default="")
@pass_factory
def system_df(ctx: Ctx, fmt):
    """Disk usage of images, sandboxes (upper layers) and volumes."""
    f = ctx.factory
    eng = f.engine()
    rows = [
        {"kind": "images", "count": len(eng.images.list()),
         "bytes": _du(consts.image_store_dir())},
        {"kind": "sandboxes", "count": len(eng.list()),
         "bytes": _du(consts.sandbox_store_dir())},
        {"kind": "volumes", "count": len(eng.db.list_volumes()),
         "bytes": _du(consts.volume_store_dir())},
        {"kind": "worktrees",
         "count": len(list((consts.data_dir() / "worktrees").glob("*/*"))),
         "bytes": _du(consts.data_dir() / "worktrees")},
    ]
    if fmt == "json":
        f.io.print(json.dumps(rows, indent=1))
        return
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for c in ("KIND", "COUNT", "SIZE"):
        t.add_column(c)
    for r in rows:
        t.add_row(r["kind"], str(r["count"]), f"{r['bytes'] / 2**20:.1f}M")
    f.io.print(t)


@system_group.command("prune")
@click.option("--volumes", "with_volumes", is_flag=True,
              help="also remove unreferenced volumes")
@pass_factory
def system_prune(ctx: Ctx, with_volumes):
    """Remove stopped sandboxes, unreferenced layers (and volumes)."""
    f = ctx.factory
    eng = f.engine()
    removed = {"sandboxes": 0, "layers": 0, "volumes": 0}
    for i in eng.list():
        if i.state != "running":
            f.orchestrator().teardown(i.name, force=True)
            removed["sandboxes"] += 1
    removed["layers"] = eng.images.prune_layers()
    if with_volumes:
        live = {i.name for i in eng.list()}
        for v in eng.db.list_volumes():
            owner = v["name"].rsplit("-", 1)[0]
            if v["name"].startswith("clawker.") and owner not in live \
                    and not v["name"].endswith(".share"):
                eng.remove_volume(v["name"])
                removed["volumes"] += 1
    f.io.success(f"pruned: {removed}")


@system_group.command("info")
@pass_factory
def system_info(ctx: Ctx):
    """Host + engine summary (docker info analog)."""
    import platform
    from .. import __version__, consts
    from ..engine.engine import detect_backend, native_bin_dir
    f = ctx.factory
    eng = f.engine()
    infos = eng.list()
    from ..gpu import GPUInventory
    inv = GPUInventory.detect()
    from ..controlplane.client import CPClient
    out = {
        "version": __version__,
        "backend": detect_backend(),
        "kernel": platform.release(),
        "native_bin": str(native_bin_dir()),
        "data_dir": str(consts.data_dir()),
        "runtime_dir": str(consts.runtime_dir()),
        "sandboxes": {
            "total": len(infos),
            "running": sum(1 for i in infos if i.state == "running"),
            "paused": sum(1 for i in infos if i.state == "paused"),
        },
        "images": len(eng.images.list()),
        "gpus": len(inv),
        "control_plane": CPClient(auto_start=False).running(),
    }
    f.io.print(json.dumps(out, indent=1))
