"""Image verbs (reference: internal/cmd/image — list/build/inspect/remove/
prune) plus the top-level `clawker build` alias."""
from __future__ import annotations

import json
import time

import click

from ..cmdutil import format_age
from .root import Ctx, cli, pass_factory


@cli.group("image")
def image_group():
    """Manage sandbox images (overlay layer stacks over hostfs)."""


@cli.command("build")
@click.option("--harness", default="", help="harness to bake (default: project agent.harness)")
@click.option("--no-cache", is_flag=True, help="rebuild the base even if fresh")
@click.option("-q", "--quiet", is_flag=True)
@pass_factory
def build_cmd(ctx: Ctx, harness, no_cache, quiet):
    """Build the project image (base + harness stages)."""
    f = ctx.factory
    cfg = f.config(require_project=True)
    from ..bundler import Builder
    builder = Builder(cfg, f.engine())
    if not quiet and f.io.is_stderr_tty():
        # live progress tree (tui components; reference: RunProgress)
        from ..tui.components import ProgressSteps
        from rich.live import Live
        steps = ProgressSteps()
        cur: list = [""]

        def progress(line: str) -> None:
            if line.startswith("building "):
                if cur[0]:
                    steps.done(cur[0])
                cur[0] = line.split(" image ", 1)[-1]
                steps.start(cur[0])
            elif cur[0]:
                steps.start(cur[0], detail=line[-60:])
            live.refresh()

        with Live(steps, console=f.io.err_console,
                  refresh_per_second=8) as live:
            name = builder.build(harness_name=harness, no_cache=no_cache,
                                 on_progress=progress)
            if cur[0]:
                steps.done(cur[0], detail="")
            live.refresh()
        f.io.print(name)
        return
    progress = None if quiet else (lambda line: f.io.eprint(f"[dim]»[/dim] {line}"))
    name = builder.build(harness_name=harness, no_cache=no_cache,
                         on_progress=progress)
    f.io.print(name)


image_group.add_command(build_cmd, "build")


@image_group.command("ls")
@click.option("--format", "fmt", default="")
@pass_factory
def image_ls(ctx: Ctx, fmt):
    """List images."""
    f = ctx.factory
    metas = f.engine().images.list()
    if fmt == "json":
        f.io.print(json.dumps([m.to_dict() for m in metas], indent=1))
        return
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for col in ("NAME", "PARENT", "LAYERS", "AGE"):
        t.add_column(col)
    now = time.time()
    for m in metas:
        t.add_row(m.name, m.parent, str(len(m.layers)),
                  format_age(now - m.created) if m.created else "-")
    f.io.print(t)


@image_group.command("inspect")
@click.argument("names", nargs=-1, required=True)
@pass_factory
def image_inspect(ctx: Ctx, names):
    f = ctx.factory
    out = [f.engine().images.get(n).to_dict() for n in names]
    f.io.print(json.dumps(out, indent=1))


@image_group.command("rm")
@click.option("-f", "--force", is_flag=True, help="remove even if in use")
@click.argument("names", nargs=-1, required=True)
@pass_factory
def image_rm(ctx: Ctx, force, names):
    f = ctx.factory
    for n in names:
        f.engine().remove_image(n, force=force)
        f.io.print(n)


image_rm.help = image_rm.help or "Remove images (docker rmi)."
cli.add_command(image_rm, "rmi")   # docker-style top-level alias


@image_group.command("save")
@click.option("-o", "--output", default="", help="output tarball (default <name>.tar.gz)")
@click.argument("name")
@pass_factory
def image_save(ctx: Ctx, output, name):
    """Export an image (manifest + layers) as a tarball (docker save
    analog — survives the tmpfs layer store across reboots)."""
    from pathlib import Path
    f = ctx.factory
    out = Path(output) if output else Path(
        name.replace("/", "_").replace(":", "_") + ".tar.gz")
    p = f.engine().images.save(name, out)
    f.io.print(str(p))


@image_group.command("load")
@click.option("--name", "rename", default="", help="register under a different name")
@click.argument("tarball", type=click.Path(exists=True))
@pass_factory
def image_load(ctx: Ctx, rename, tarball):
    """Import a tarball written by `image save`; existing layers are
    deduplicated by content id."""
    f = ctx.factory
    meta = f.engine().images.load(tarball, rename=rename)
    f.io.print(f"{meta.name}  layers={len(meta.layers)}")


@image_group.command("history")
@click.argument("name")
@pass_factory
def image_history(ctx: Ctx, name):
    """Layer chain of an image, top-most first (docker history analog)."""
    import subprocess
    f = ctx.factory
    store = f.engine().images
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for c in ("IMAGE", "LAYER", "SIZE", "CREATED"):
        t.add_column(c)
    now = time.time()
    for meta in store._image_chain(name):
        for lid in reversed(meta.layers):
            du = subprocess.run(["du", "-sh", str(store.layer_path(lid))],
                                capture_output=True, text=True)
            size = du.stdout.split()[0] if du.returncode == 0 else "?"
            t.add_row(meta.name, lid[:12], size,
                      format_age(now - meta.created) if meta.created else "-")
    t.add_row("hostfs", "(host rootfs)", "-", "-")
    f.io.print(t)


@image_group.command("prune")
@pass_factory
def image_prune(ctx: Ctx):
    """Remove layers referenced by no image."""
    f = ctx.factory
    n = f.engine().images.prune_layers()
    f.io.eprint(f"removed {n} unreferenced layer(s)")


@cli.group("harness")
def harness_group():
    """Agent harness bundles (claude, codex, ...)."""


@harness_group.command("list")
@pass_factory
def harness_list(ctx: Ctx):
    from ..bundle import list_harnesses, load_harness
    f = ctx.factory
    cfg = f.config()
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for c in ("NAME", "CMD", "STACKS", "EGRESS RULES"):
        t.add_column(c)
    for name in list_harnesses(cfg.project_root):
        try:
            h = load_harness(name, cfg.project_root)
            t.add_row(name, " ".join(h.cmd)[:40], ",".join(h.stacks) or "-",
                      str(len(h.egress)))
        except Exception:
            t.add_row(name, "?", "?", "?")
    f.io.print(t)


@harness_group.command("show")
@click.argument("name")
@pass_factory
def harness_show(ctx: Ctx, name):
    from ..bundle import load_harness
    from ..storage.store import to_plain
    h = load_harness(name, ctx.factory.config().project_root)
    ctx.factory.io.print(json.dumps(to_plain(h), indent=1))


@cli.group("stack")
def stack_group():
    """Language/toolchain stacks baked into base images."""


@stack_group.command("list")
@pass_factory
def stack_list(ctx: Ctx):
    from ..bundle.loader import ASSETS, load_stack
    f = ctx.factory
    names = sorted(p.stem for p in (ASSETS / "stacks").glob("*.yaml"))
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for c in ("NAME", "PACKAGES", "DESCRIPTION"):
        t.add_column(c)
    for n in names:
        s = load_stack(n)
        t.add_row(n, ",".join(s.packages) or "-", s.description)
    f.io.print(t)
