"""Prompt + bundle verbs (reference: internal/cmd/prompt and
internal/cmd/bundle — named prompt management and bundle install;
the git-fetch tier is pointless on an air-gapped node, so `bundle
install` takes a local directory)."""
from __future__ import annotations

import shutil
import time

import click

from .. import consts
from ..errors import ClawkerError, NotFoundError
from .root import Ctx, cli, pass_factory


def prompts_dir():
    d = consts.data_dir() / "prompts"
    d.mkdir(parents=True, exist_ok=True)
    return d


def resolve_prompt(name_or_path: str):
    """A stored prompt name, or a literal file path."""
    from pathlib import Path
    p = prompts_dir() / f"{name_or_path}.md"
    if p.is_file():
        return p
    lit = Path(name_or_path)
    if lit.is_file():
        return lit
    raise NotFoundError(f"no prompt named or at: {name_or_path}")


@cli.group("prompt")
def prompt_group():
    """Named one-shot prompts for agent fleets."""


@prompt_group.command("add")
@click.argument("name")
@click.argument("source", type=click.Path(exists=True))
@pass_factory
def prompt_add(ctx: Ctx, name, source):
    """Store a prompt file under NAME (use with `fleet up --prompt NAME`)."""
    dst = prompts_dir() / f"{name}.md"
    shutil.copyfile(source, dst)
    ctx.factory.io.success(f"prompt '{name}' stored ({dst})")


@prompt_group.command("list")
@pass_factory
def prompt_list(ctx: Ctx):
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for c in ("NAME", "SIZE", "FIRST LINE"):
        t.add_column(c)
    for p in sorted(prompts_dir().glob("*.md")):
        first = (p.read_text().splitlines() or [""])[0][:60]
        t.add_row(p.stem, f"{p.stat().st_size}B", first)
    ctx.factory.io.print(t)


@prompt_group.command("show")
@click.argument("name")
@pass_factory
def prompt_show(ctx: Ctx, name):
    ctx.factory.io.print(resolve_prompt(name).read_text())


@prompt_group.command("rm")
@click.argument("name")
@pass_factory
def prompt_rm(ctx: Ctx, name):
    p = prompts_dir() / f"{name}.md"
    if not p.is_file():
        raise NotFoundError(f"no prompt named: {name}")
    p.unlink()
    ctx.factory.io.success(f"removed prompt '{name}'")


@cli.group("bundle")
def bundle_group():
    """Harness/stack bundles: install (dir or git), list, gc."""


@bundle_group.command("install")
@click.argument("source")
@click.option("--kind", type=click.Choice(["harnesses", "stacks"]),
              default="harnesses", show_default=True)
@click.option("--name", default="", help="override the installed name")
@click.option("--ref", default="", help="git branch/tag (git sources)")
@pass_factory
def bundle_install(ctx: Ctx, source, kind, name, ref):
    """Install a bundle into the user tier
    (~/.config/clawker/<kind>/<name>); it then resolves above the
    embedded floor. SOURCE is a local directory or a git URL
    (https://, git@, file://, or a local repo path ending in .git) —
    the fetch pipeline caches by value and strips escaping symlinks
    (reference: internal/bundle install.go)."""
    from pathlib import Path
    if (source.startswith(("http://", "https://", "git@", "file://",
                           "ssh://")) or source.endswith(".git")):
        from ..bundle.install import install_from_git
        k, n = install_from_git(source, ref=ref, name=name)
        ctx.factory.io.success(f"installed {k[:-2]} bundle '{n}' from git")
        return
    src = Path(source)
    if not src.is_dir():
        raise ClawkerError(f"not a directory or git URL: {source}")
    manifest = None
    for cand in ("harness.yaml", "manifest.yaml", "stack.yaml"):
        if (src / cand).is_file():
            manifest = cand
            break
    if manifest is None and kind == "stacks":
        raise ClawkerError(f"{src}: no stack.yaml/manifest.yaml found")
    if manifest is None:
        raise ClawkerError(f"{src}: no harness.yaml/manifest.yaml found")
    dst = consts.config_dir() / kind / src.name
    if dst.exists():
        shutil.rmtree(dst)
    # symlink-escape sanitization (reference: bundle install pipeline)
    def _no_symlinks(d, names):
        return [n for n in names if (Path(d) / n).is_symlink()]
    shutil.copytree(src, dst, ignore=_no_symlinks)
    (dst / ".installed").write_text(str(time.time()))
    ctx.factory.io.success(f"installed {kind[:-2]} bundle '{src.name}' -> {dst}")


@bundle_group.command("gc")
@click.option("--dry-run", is_flag=True)
@pass_factory
def bundle_gc(ctx: Ctx, dry_run):
    """Remove installed bundles + cache entries no registered project
    declares (reference: bundle GC against declaration roots)."""
    from ..bundle.install import gc
    removed = gc(dry_run=dry_run)
    n = sum(len(v) for v in removed.values())
    verb = "would remove" if dry_run else "removed"
    ctx.factory.io.print(f"{verb} {n} item(s): "
                         + ", ".join(f"{k}={v}" for k, v in removed.items() if v)
                         if n else "nothing to collect")


@bundle_group.command("list")
@pass_factory
def bundle_list(ctx: Ctx):
    from ..bundle import list_harnesses
    f = ctx.factory
    cfg = f.config()
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for c in ("KIND", "NAME", "TIER"):
        t.add_column(c)
    user_dir = consts.config_dir() / "harnesses"
    proj_dir = (cfg.project_root / consts.PROJECT_DIR_NAME / "harnesses"
                if cfg.project_root else None)
    for name in list_harnesses(cfg.project_root):
        tier = "embedded"
        if proj_dir and (proj_dir / name).exists():
            tier = "project"
        elif (user_dir / name).exists() or (user_dir / f"{name}.yaml").exists():
            tier = "user"
        t.add_row("harness", name, tier)
    f.io.print(t)


# reference parity: the bundle tier is also exposed as `plugin` with a
# `skill` alias (internal/cmd plugin group, alias skill — SURVEY.md §2.1)
cli.add_command(bundle_group, "plugin")
cli.add_command(bundle_group, "skill")
