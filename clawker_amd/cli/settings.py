"""Settings + alias verbs (reference: internal/cmd/settings `settings edit`
storeui TUI + internal/cmd/alias; here: get/set/edit via $EDITOR)."""
from __future__ import annotations

import json
import os
import subprocess

import click

from .. import consts
from ..config.config import load_settings
from ..errors import ClawkerError
from ..storage.store import to_plain
from .root import Ctx, cli, pass_factory


@cli.group("settings")
def settings_group():
    """Global settings (settings.yaml)."""


@settings_group.command("get")
@click.argument("path", required=False)
@pass_factory
def settings_get(ctx: Ctx, path):
    store = load_settings()
    if path:
        ctx.factory.io.print(json.dumps(store.get_path(path)))
    else:
        ctx.factory.io.print(json.dumps(to_plain(store.get()), indent=1))


@settings_group.command("set")
@click.argument("path")
@click.argument("value")
@pass_factory
def settings_set(ctx: Ctx, path, value):
    """Set a settings field (value parsed as YAML/JSON scalar)."""
    import yaml
    store = load_settings()
    parsed = yaml.safe_load(value)
    layer = store.set(path, parsed)
    store.write()
    ctx.factory.io.success(f"{path} = {parsed!r} (layer: {layer})")


@settings_group.command("edit")
@click.option("-i", "--interactive", is_flag=True,
              help="field-by-field editor instead of $EDITOR")
@pass_factory
def settings_edit(ctx: Ctx, interactive):
    """Edit settings.yaml ($EDITOR, or -i for the field editor)."""
    if interactive:
        from ..storeui import edit_store
        n = edit_store(load_settings(), ctx.factory.io)
        ctx.factory.io.success(f"{n} field(s) updated")
        return
    path = consts.config_dir() / consts.SETTINGS_BASENAME
    path.parent.mkdir(parents=True, exist_ok=True)
    if not path.exists():
        store = load_settings()
        import yaml
        path.write_text(yaml.safe_dump(to_plain(store.get()), sort_keys=False))
    editor = os.environ.get("EDITOR") or os.environ.get("VISUAL") or "nano"
    subprocess.run([editor, str(path)], check=False)
    load_settings().get()   # validate
    ctx.factory.io.success(f"saved {path}")


@cli.group("alias")
def alias_group():
    """User-defined command aliases (stored in settings.yaml)."""


@alias_group.command("set")
@click.argument("name")
@click.argument("expansion")
@pass_factory
def alias_set(ctx: Ctx, name, expansion):
    store = load_settings()
    aliases = dict(store.get().aliases)
    aliases[name] = expansion
    store.set("aliases", aliases)
    store.write()
    ctx.factory.io.success(f"alias {name} = {expansion}")


@alias_group.command("list")
@pass_factory
def alias_list(ctx: Ctx):
    for k, v in sorted(load_settings().get().aliases.items()):
        ctx.factory.io.print(f"{k} = {v}")


@alias_group.command("delete")
@click.argument("name")
@pass_factory
def alias_delete(ctx: Ctx, name):
    store = load_settings()
    aliases = dict(store.get().aliases)
    if name not in aliases:
        raise ClawkerError(f"no such alias: {name}")
    del aliases[name]
    store.set("aliases", aliases)
    store.write()
    ctx.factory.io.success(f"deleted alias {name}")


@cli.command("version")
@pass_factory
def version_cmd(ctx: Ctx):
    """Print the clawker-amd version."""
    from .. import __version__
    from ..engine.engine import detect_backend
    ctx.factory.io.print(f"clawker-amd {__version__} "
                         f"(isolation backend: {detect_backend()})")


@cli.command("completion")
@click.argument("shell", type=click.Choice(["bash", "zsh", "fish"]))
@pass_factory
def completion_cmd(ctx: Ctx, shell):
    """Print the shell-completion script (eval or source it: e.g.
    `source <(clawker completion bash)`)."""
    from click.shell_completion import get_completion_class
    from .root import cli as root_cli
    cls = get_completion_class(shell)
    comp = cls(root_cli, {}, "clawker", "_CLAWKER_COMPLETE")
    ctx.factory.io.print(comp.source())
