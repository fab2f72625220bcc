"""Worktree verbs (reference: internal/cmd/worktree add/list/prune/remove)."""
from __future__ import annotations

import json

import click

from ..cmdutil import format_age
from ..project import worktrees as wt
from .root import Ctx, cli, pass_factory


@cli.group("worktree")
def worktree_group():
    """Git worktrees for parallel agent fan-out (one pinned GPU each)."""


@worktree_group.command("add")
@click.argument("branch")
@click.option("--base", default="", help="start point for a new branch")
@pass_factory
def worktree_add(ctx: Ctx, branch, base):
    f = ctx.factory
    cfg = f.config(require_project=True)
    w = wt.setup_worktree(cfg, branch, base)
    f.io.success(f"worktree '{branch}' at {w.path}")
    f.io.print(str(w.path))


@worktree_group.command("list")
@click.option("--format", "fmt", default="")
@pass_factory
def worktree_list(ctx: Ctx, fmt):
    f = ctx.factory
    cfg = f.config(require_project=True)
    rows = wt.list_worktrees(cfg)
    if fmt == "json":
        f.io.print(json.dumps(rows, indent=1))
        return
    import time
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for c in ("BRANCH", "PATH", "STATUS", "AGE"):
        t.add_column(c)
    now = time.time()
    for r in rows:
        t.add_row(r["branch"], r["path"], r["status"],
                  format_age(now - r["created"]) if r["created"] else "-")
    f.io.print(t)


@worktree_group.command("remove")
@click.argument("branch")
@click.option("-f", "--force", is_flag=True, help="discard local changes")
@pass_factory
def worktree_remove(ctx: Ctx, branch, force):
    f = ctx.factory
    cfg = f.config(require_project=True)
    wt.remove_worktree(cfg, branch, force=force)
    f.io.success(f"removed worktree '{branch}'")


@worktree_group.command("prune")
@pass_factory
def worktree_prune(ctx: Ctx):
    f = ctx.factory
    cfg = f.config(require_project=True)
    removed = wt.prune_worktrees(cfg)
    f.io.eprint(f"pruned {len(removed)} stale worktree entr(ies): {removed}")
