"""clawker CLI root (reference: internal/cmd/root — Cobra root with 20+
command groups plus Docker-style top-level aliases; here click).

Layout mirrors the reference's verb groups:
  clawker run/create/start/stop/ps/rm/exec/logs/attach/wait/stats/kill  (container aliases)
  clawker container <verb>      (the full group)
  clawker image build/ls/rm/inspect/prune
  clawker volume ls/rm
  clawker project init/list/info/remove ; clawker init
  clawker worktree add/list/remove/prune
  clawker firewall status/list/add/remove/enable/disable/bypass/reload
  clawker cp up/down/status/agents      (control plane)
  clawker monitor up/down/status ; clawker gpus
  clawker settings edit/get/set ; clawker alias
  clawker version
"""
from __future__ import annotations

import sys

import click

from .. import __version__, consts, logger
from ..cmdutil import Factory
from ..errors import ClawkerError, FlagError, SilentError


class Ctx:
    def __init__(self):
        self.factory = Factory()


pass_factory = click.make_pass_decorator(Ctx, ensure=True)


@click.group(context_settings={"help_option_names": ["-h", "--help"]})
@click.version_option(__version__, prog_name="clawker")
def cli():
    """MI355X-native agent-in-container orchestrator.

    Runs AI coding-agent harnesses in isolated sandboxes with per-agent
    GPU pinning, deny-by-default egress and live ROCm telemetry."""
    logger.setup(consts.log_dir() / "clawker.log")


def expand_user_alias(argv: list[str]) -> list[str]:
    """User-defined alias expansion with $1..$N / $@ substitution
    (reference: internal/cmd/root/useraliases.go). Only the first token
    is eligible and built-ins always win."""
    if not argv or argv[0].startswith("-") or argv[0] in cli.commands:
        return argv
    try:
        from ..config.config import load_settings
        aliases = load_settings().get().aliases
    except Exception:
        return argv
    expansion = aliases.get(argv[0])
    if not expansion:
        return argv
    rest = argv[1:]
    out: list[str] = []
    used = set()
    import shlex
    for tok in shlex.split(str(expansion)):
        if tok == "$@":
            out.extend(rest)
            used.update(range(len(rest)))
        elif len(tok) >= 2 and tok[0] == "$" and tok[1:].isdigit():
            idx = int(tok[1:]) - 1
            if idx >= len(rest):
                raise FlagError(f"alias '{argv[0]}' needs argument ${tok[1:]}")
            out.append(rest[idx])
            used.add(idx)
        else:
            out.append(tok)
    out.extend(a for i, a in enumerate(rest) if i not in used and "$" not in expansion)
    return out


def main() -> int:
    from . import container, doctor, firewall, fleet, image, monitor, project, prompt, settings, system, cp, volume, worktree  # noqa
    try:       # once-per-version changelog teaser (interactive runs only)
        if sys.stderr.isatty():
            from ..iostreams import IOStreams
            from ..update import maybe_show_teaser
            maybe_show_teaser(IOStreams())
    except Exception:
        pass
    try:
        cli(args=expand_user_alias(sys.argv[1:]), standalone_mode=False)
        return 0
    except click.exceptions.Abort:
        return 130
    except click.exceptions.ClickException as e:
        e.show()
        return e.exit_code
    except FlagError as e:
        click.echo(f"error: {e}", err=True)
        return 2
    except SilentError as e:
        return e.exit_code
    except ClawkerError as e:
        msg = e.user_message()
        if msg:
            click.echo(f"clawker: {msg}", err=True)
        return e.exit_code
    except KeyboardInterrupt:
        return 130


# import groups at module load so `clawker --help` lists them
from . import container, doctor, firewall, fleet, image, monitor, project, prompt, settings, system, cp, volume, worktree  # noqa: E402,F401

if __name__ == "__main__":
    # re-enter through the canonical module path so command registration
    # (which imports clawker_amd.cli.root) targets THIS cli object
    from clawker_amd.cli.root import main as _main
    sys.exit(_main())
