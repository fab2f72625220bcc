"""Config facade over Store[Project] + Store[Settings].

Reference: internal/config/config.go — the Config interface commands use
(never importing storage directly); path accessors; project discovery.
"""
from __future__ import annotations

import os
import re
from pathlib import Path

from .. import consts
from ..errors import ClawkerError
from ..storage import Layer, Store, discover_project_layers
from .schema import Project, Settings

_SLUG_RE = re.compile(r"[^a-z0-9_-]+")


def slugify(name: str) -> str:
    s = _SLUG_RE.sub("-", name.lower()).strip("-")
    return s or "project"


def settings_layers() -> list[Layer]:
    return [Layer(name="settings", path=consts.config_dir() / consts.SETTINGS_BASENAME)]


def load_settings() -> Store[Settings]:
    return Store(Settings, settings_layers())


class Config:
    """One facade: project store (layered, walk-up) + settings store."""

    def __init__(self, project_store: Store[Project], settings_store: Store[Settings],
                 project_root: Path | None):
        self._project = project_store
        self._settings = settings_store
        self.project_root = project_root

    # -- typed views ---------------------------------------------------------
    @property
    def project(self) -> Project:
        p = self._project.get()
        if not p.project and self.project_root is not None:
            p.project = slugify(self.project_root.name)
        return p

    @property
    def settings(self) -> Settings:
        return self._settings.get()

    @property
    def project_store(self) -> Store[Project]:
        return self._project

    @property
    def settings_store(self) -> Store[Settings]:
        return self._settings

    # -- identity ------------------------------------------------------------
    @property
    def project_slug(self) -> str:
        return self.project.project

    def sandbox_name(self, agent: str) -> str:
        """clawker.<project>.<agent> (reference: docker/names.go)."""
        return f"{consts.SANDBOX_NAME_PREFIX}{self.project_slug}.{agent}"

    def image_name(self, harness: str | None = None) -> str:
        h = harness or self.project.agent.harness or "default"
        return f"clawker-{self.project_slug}:{h}"

    def base_image_name(self) -> str:
        return f"clawker-{self.project_slug}:base"

    # -- paths ---------------------------------------------------------------
    def workspace_path(self) -> Path:
        ws = self.project.workspace.path
        if ws:
            return Path(ws).expanduser().resolve()
        if self.project_root is None:
            raise ClawkerError("no project root: run inside a project or pass --project")
        return self.project_root


def load_config(cwd: Path | None = None, require_project: bool = False) -> Config:
    """Discover project layers by walk-up from cwd and build the facade."""
    cwd = (cwd or Path.cwd()).resolve()
    paths = discover_project_layers(cwd)
    layers: list[Layer] = []
    roots: list[Path] = []
    for p in paths:
        name = "project-local" if p.name == consts.PROJECT_LOCAL_BASENAME else "project"
        if name == "project":
            name = f"project:{p.parent}"
        layers.append(Layer(name=name, path=p))
        base = p.parent
        if base.name == consts.PROJECT_DIR_NAME:
            base = base.parent
        if base not in roots:
            roots.append(base)
    # the project ROOT is registry-resolved when possible (reference:
    # project identity resolution, resolve.go); otherwise the OUTERMOST
    # discovered config dir — nested config files inside a project are
    # walk-up override layers, not new roots
    root: Path | None = None
    if roots:
        try:
            from ..project import ProjectRegistry
            entry = ProjectRegistry().resolve_by_path(cwd)
        except Exception:
            entry = None
        if entry is not None and Path(entry.root) in roots:
            root = Path(entry.root)
        else:
            root = roots[0]     # outermost (discover order: farthest first)
    if require_project and root is None:
        raise ClawkerError(
            "no clawker project found (run `clawker init` at your project root)")
    project_store = Store(Project, layers)
    return Config(project_store, load_settings(), root)


def ensure_dirs() -> None:
    for d in (consts.config_dir(), consts.data_dir(), consts.state_dir(),
              consts.log_dir(), consts.image_store_dir(), consts.sandbox_store_dir(),
              consts.volume_store_dir()):
        d.mkdir(parents=True, exist_ok=True)
    rd = consts.runtime_dir()
    rd.mkdir(parents=True, exist_ok=True)
    if os.geteuid() == 0:
        os.chmod(rd, 0o700)
