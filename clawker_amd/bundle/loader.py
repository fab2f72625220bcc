"""Harness/stack bundle resolution.

Reference: internal/bundle (three-tier resolution: embedded floor assets /
loose local dirs / installed cache — resolver.go:73) + the shipped floor
content in internal/bundle/assets (harnesses claude + codex, 8 language
stacks). This node has no network, so the "installed bundle cache" tier is
omitted; resolution is: project-local dir > user config dir > embedded
floor (clawker_amd/bundle/assets).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from pathlib import Path

import yaml

from .. import consts
from ..config.schema import EgressRule
from ..errors import NotFoundError
from ..storage import materialize
from .versions import VersionSpec

ASSETS = Path(__file__).resolve().parent / "assets"


@dataclass
class BuildStep:
    run: str = ""
    best_effort: bool = False   # ignore failure (e.g. network installs on
                                # an air-gapped node)


@dataclass
class Harness:
    name: str = ""
    description: str = ""
    cmd: list[str] = field(default_factory=list)
    # how to hand a one-shot prompt to the harness; "@PROMPT_FILE@" is
    # replaced with the in-sandbox prompt path (fleet --prompt)
    prompt_cmd: list[str] = field(default_factory=list)
    user: str = "agent"
    # how the harness CLI's version is resolved at build time
    # (reference: versions.go; "@VERSION@" in install steps expands)
    version: VersionSpec = field(default_factory=VersionSpec)
    stacks: list[str] = field(default_factory=list, metadata={"merge": "union"})
    install: list[BuildStep] = field(default_factory=list)
    env: dict = field(default_factory=dict)
    config_volumes: list[str] = field(default_factory=list)
    managed_prompt_path: str = ""
    egress: list[EgressRule] = field(default_factory=list, metadata={"merge": "union"})
    seeds: dict = field(default_factory=dict)      # in-image path -> content
    staging: list[dict] = field(default_factory=list)   # host-state staging
    post_init: str = ""     # script body (InitPlan)
    pre_run: str = ""       # script body (BootPlan)


@dataclass
class Stack:
    name: str = ""
    description: str = ""
    packages: list[str] = field(default_factory=list)
    steps: list[BuildStep] = field(default_factory=list)
    env: dict = field(default_factory=dict)


def _search_dirs(kind: str, project_root: Path | None) -> list[Path]:
    dirs = []
    if project_root is not None:
        dirs.append(project_root / consts.PROJECT_DIR_NAME / kind)
    dirs.append(consts.config_dir() / kind)
    dirs.append(ASSETS / kind)
    return dirs


# mtime-keyed parse cache: the claude floor manifest alone costs ~24 ms
# of PyYAML per load, and the hot path loads a harness twice per sandbox
# create (orchestrator + firewall floor composition)
_yaml_cache: dict[tuple, tuple[float, dict]] = {}


def _load_yaml(kind: str, name: str, project_root: Path | None) -> dict:
    for d in _search_dirs(kind, project_root):
        for candidate in (d / name / f"{kind[:-2] if kind.endswith('es') else kind}.yaml",
                          d / f"{name}.yaml",
                          d / name / "manifest.yaml"):
            if candidate.is_file():
                key = (kind, str(candidate))
                mtime = candidate.stat().st_mtime
                hit = _yaml_cache.get(key)
                if hit is not None and hit[0] == mtime:
                    return dict(hit[1])
                data = yaml.safe_load(candidate.read_text()) or {}
                data.setdefault("name", name)
                _yaml_cache[key] = (mtime, dict(data))
                return data
    raise NotFoundError(f"{kind[:-2] if kind.endswith('es') else kind} not found: {name}")


def load_harness(name: str, project_root: Path | None = None) -> Harness:
    data = _load_yaml("harnesses", name, project_root)
    # normalize install/steps: strings become BuildStep
    inst = data.get("install") or []
    data["install"] = [{"run": s} if isinstance(s, str) else s for s in inst]
    return materialize(Harness, data)


def load_stack(name: str, project_root: Path | None = None) -> Stack:
    data = _load_yaml("stacks", name, project_root)
    steps = data.get("steps") or []
    data["steps"] = [{"run": s} if isinstance(s, str) else s for s in steps]
    return materialize(Stack, data)


def list_harnesses(project_root: Path | None = None) -> list[str]:
    names: set[str] = set()
    for d in _search_dirs("harnesses", project_root):
        if d.is_dir():
            for p in d.iterdir():
                if p.is_dir() and any((p / n).is_file()
                                      for n in ("harness.yaml", "manifest.yaml")):
                    names.add(p.name)
                elif p.suffix == ".yaml":
                    names.add(p.stem)
    return sorted(names)


def harness_egress_floor(harness: Harness, project_rules: list[EgressRule]) -> list[EgressRule]:
    """Egress composition: harness floor ∪ project rules (reference:
    bundler/egress.go:22 EgressRules)."""
    seen: set[str] = set()
    out: list[EgressRule] = []
    for r in list(harness.egress) + list(project_rules):
        if r.key() not in seen:
            seen.add(r.key())
            out.append(r)
    return out
