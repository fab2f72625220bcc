"""Factory DI contract for commands (reference: internal/cmdutil/factory.go
— a pure struct of lazily-wired closures; commands never construct heavy
deps themselves, tests inject fakes)."""
from __future__ import annotations

from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Callable, Optional

from .config import Config, load_config
from .iostreams import IOStreams
from .prompter import Prompter


@dataclass
class Factory:
    io: IOStreams = field(default_factory=IOStreams)
    cwd: Path = field(default_factory=Path.cwd)
    _config: Optional[Config] = None
    _engine: Any = None
    _orchestrator: Any = None
    _prompter: Optional[Prompter] = None
    _cp: Any = None

    # overridable constructors (tests swap these; reference: runF hook)
    config_fn: Callable[["Factory"], Config] | None = None
    engine_fn: Callable[["Factory"], Any] | None = None
    orchestrator_fn: Callable[["Factory"], Any] | None = None
    cp_fn: Callable[["Factory"], Any] | None = None

    def config(self, require_project: bool = False) -> Config:
        if self._config is None:
            if self.config_fn:
                self._config = self.config_fn(self)
            else:
                self._config = load_config(self.cwd)
        if require_project and self._config.project_root is None:
            from .errors import ClawkerError
            raise ClawkerError(
                "no clawker project found (run `clawker init` at your project root)")
        return self._config

    def engine(self):
        if self._engine is None:
            if self.engine_fn:
                self._engine = self.engine_fn(self)
            else:
                from .engine import Engine
                self._engine = Engine()
        return self._engine

    def orchestrator(self):
        if self._orchestrator is None:
            if self.orchestrator_fn:
                self._orchestrator = self.orchestrator_fn(self)
            else:
                from .orchestrator import Orchestrator
                self._orchestrator = Orchestrator(self.config(), self.engine())
        return self._orchestrator

    def prompter(self) -> Prompter:
        if self._prompter is None:
            self._prompter = Prompter(self.io)
        return self._prompter

    def controlplane(self):
        """CP admin client (lazy; starts the daemon on demand)."""
        if self._cp is None:
            if self.cp_fn:
                self._cp = self.cp_fn(self)
            else:
                from .controlplane.client import CPClient
                self._cp = CPClient()
        return self._cp


def resolve_sandbox_name(f: Factory, name_or_agent: str) -> str:
    """Accept either a full sandbox name (clawker.<proj>.<agent>) or a bare
    agent name resolved against the current project (reference: cmdutil
    container name resolution)."""
    from . import consts
    if name_or_agent.startswith(consts.SANDBOX_NAME_PREFIX):
        return name_or_agent
    cfg = f.config(require_project=True)
    return cfg.sandbox_name(name_or_agent)


def format_age(seconds: float) -> str:
    if seconds < 60:
        return f"{int(seconds)}s"
    if seconds < 3600:
        return f"{int(seconds // 60)}m"
    if seconds < 86400:
        return f"{seconds / 3600:.1f}h"
    return f"{seconds / 86400:.1f}d"
