"""Project registry: registry.yaml in the XDG data dir.

Reference: internal/project (registry_schema.go:11 ProjectEntry, :18
WorktreeEntry; identity resolution resolve.go; worktree lifecycle
worktree_service.go; runtime health enrichment manager.go).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from pathlib import Path

from .. import consts
from ..errors import ClawkerError, NotFoundError
from ..storage import Layer, Store


@dataclass
class ProjectEntry:
    name: str = ""
    root: str = ""
    created: float = 0.0


@dataclass
class WorktreeEntry:
    project: str = ""
    branch: str = ""
    path: str = ""
    base: str = ""
    created: float = 0.0


@dataclass
class RegistrySchema:
    version: int = 1
    projects: list[ProjectEntry] = field(default_factory=list)
    worktrees: list[WorktreeEntry] = field(default_factory=list)


class ProjectRegistry:
    def __init__(self, path: Path | None = None):
        self.path = path or (consts.data_dir() / consts.REGISTRY_BASENAME)
        self.store: Store[RegistrySchema] = Store(
            RegistrySchema, [Layer(name="registry", path=self.path)])

    def _data(self) -> RegistrySchema:
        return self.store.get()

    # -- projects --------------------------------------------------------------
    def register(self, name: str, root: Path) -> ProjectEntry:
        data = self._data()
        for p in data.projects:
            if p.name == name:
                if Path(p.root) != root:
                    raise ClawkerError(
                        f"project '{name}' already registered at {p.root}")
                return p
        entry = ProjectEntry(name=name, root=str(root), created=time.time())
        projects = [p.__dict__ for p in data.projects] + [entry.__dict__]
        self.store.set("projects", projects, layer="registry")
        self.store.write()
        return entry

    def unregister(self, name: str) -> None:
        data = self._data()
        remaining = [p.__dict__ for p in data.projects if p.name != name]
        if len(remaining) == len(data.projects):
            raise NotFoundError(f"project not registered: {name}")
        self.store.set("projects", remaining, layer="registry")
        self.store.set("worktrees",
                       [w.__dict__ for w in data.worktrees if w.project != name],
                       layer="registry")
        self.store.write()

    def list_projects(self) -> list[ProjectEntry]:
        return self._data().projects

    def get(self, name: str) -> ProjectEntry:
        for p in self._data().projects:
            if p.name == name:
                return p
        raise NotFoundError(f"project not registered: {name}")

    def resolve_by_path(self, path: Path) -> ProjectEntry | None:
        """Innermost registered project containing path."""
        path = path.resolve()
        best: ProjectEntry | None = None
        for p in self._data().projects:
            root = Path(p.root)
            try:
                path.relative_to(root)
            except ValueError:
                continue
            if best is None or len(str(root)) > len(best.root):
                best = p
        return best

    # -- worktrees -------------------------------------------------------------
    def add_worktree(self, entry: WorktreeEntry) -> None:
        data = self._data()
        wts = [w.__dict__ for w in data.worktrees
               if not (w.project == entry.project and w.branch == entry.branch)]
        entry.created = entry.created or time.time()
        wts.append(entry.__dict__)
        self.store.set("worktrees", wts, layer="registry")
        self.store.write()

    def remove_worktree(self, project: str, branch: str) -> None:
        data = self._data()
        wts = [w.__dict__ for w in data.worktrees
               if not (w.project == project and w.branch == branch)]
        self.store.set("worktrees", wts, layer="registry")
        self.store.write()

    def list_worktrees(self, project: str | None = None) -> list[WorktreeEntry]:
        wts = self._data().worktrees
        if project:
            wts = [w for w in wts if w.project == project]
        return wts
