"""Git worktree fan-out: the N-parallel-agents primitive.

Reference: internal/git (git.go:191 SetupWorktree, :356 RemoveWorktree,
:392 ListWorktrees) + project/worktree_service.go + SURVEY.md A.2. The
MI355X shape: N worktree sandboxes, one pinned GPU each — the allocator
handles the pinning; this module handles the git side via the git CLI
(go-git equivalent; this image ships git 2.34).
"""
from __future__ import annotations

import re
import subprocess
import time
from dataclasses import dataclass
from pathlib import Path

from .. import consts
from ..config import Config
from ..errors import ClawkerError
from ..logger import get as get_logger
from .registry import ProjectRegistry, WorktreeEntry

log = get_logger("worktree")


@dataclass
class Worktree:
    branch: str
    path: Path
    base: str = ""

    @property
    def safe_name(self) -> str:
        return re.sub(r"[^a-zA-Z0-9_-]+", "-", self.branch).strip("-")


def _git(repo: Path, *args: str, check: bool = True) -> subprocess.CompletedProcess:
    r = subprocess.run(["git", "-C", str(repo), *args],
                       capture_output=True, text=True)
    if check and r.returncode != 0:
        raise ClawkerError(f"git {' '.join(args)}: {r.stderr.strip()}")
    return r


def is_git_repo(path: Path) -> bool:
    return _git(path, "rev-parse", "--git-dir", check=False).returncode == 0


def worktrees_dir(cfg: Config) -> Path:
    return consts.data_dir() / "worktrees" / cfg.project_slug


def setup_worktree(cfg: Config, branch: str, base: str = "") -> Worktree:
    """Create (or reuse) a worktree for `branch`; new branch from `base`
    (default: current HEAD) when it does not exist."""
    root = cfg.project_root
    if root is None or not is_git_repo(root):
        raise ClawkerError("worktrees need a git repository project root")
    wt_path = worktrees_dir(cfg) / re.sub(r"[^a-zA-Z0-9_.-]+", "-", branch)
    if wt_path.exists():
        # verify it is still a registered git worktree
        r = _git(root, "worktree", "list", "--porcelain", check=False)
        if str(wt_path) in r.stdout:
            return Worktree(branch=branch, path=wt_path, base=base)
        raise ClawkerError(
            f"worktree dir exists but is not a git worktree: {wt_path} "
            f"(run `clawker worktree prune`)")
    wt_path.parent.mkdir(parents=True, exist_ok=True)
    branch_exists = _git(root, "rev-parse", "--verify", "--quiet",
                         f"refs/heads/{branch}", check=False).returncode == 0
    if branch_exists:
        _git(root, "worktree", "add", str(wt_path), branch)
    else:
        start = base or "HEAD"
        _git(root, "worktree", "add", "-b", branch, str(wt_path), start)
    log.info("worktree_created", branch=branch, path=str(wt_path))
    reg = ProjectRegistry()
    reg.add_worktree(WorktreeEntry(
        project=cfg.project_slug, branch=branch, path=str(wt_path),
        base=base, created=time.time()))
    return Worktree(branch=branch, path=wt_path, base=base)


def ensure_worktree(cfg: Config, flag: str) -> Worktree:
    """Parse the --worktree flag BRANCH[:BASE] and ensure the worktree."""
    branch, _, base = flag.partition(":")
    if not branch:
        raise ClawkerError("--worktree needs BRANCH[:BASE]")
    return setup_worktree(cfg, branch, base)


def remove_worktree(cfg: Config, branch: str, force: bool = False) -> None:
    root = cfg.project_root
    if root is None:
        raise ClawkerError("not in a project")
    wt_path = worktrees_dir(cfg) / re.sub(r"[^a-zA-Z0-9_.-]+", "-", branch)
    args = ["worktree", "remove", str(wt_path)]
    if force:
        args.append("--force")
    _git(root, *args)
    ProjectRegistry().remove_worktree(cfg.project_slug, branch)
    log.info("worktree_removed", branch=branch)


def list_worktrees(cfg: Config) -> list[dict]:
    """Registered worktrees enriched with live git status
    (reference: worktree_service.go classification for prune)."""
    reg = ProjectRegistry()
    root = cfg.project_root
    live: set[str] = set()
    if root is not None and is_git_repo(root):
        r = _git(root, "worktree", "list", "--porcelain", check=False)
        for line in r.stdout.splitlines():
            if line.startswith("worktree "):
                live.add(line.split(" ", 1)[1])
    out = []
    for w in reg.list_worktrees(cfg.project_slug):
        out.append({
            "branch": w.branch, "path": w.path, "base": w.base,
            "created": w.created,
            "status": "ok" if w.path in live else "stale",
        })
    return out


def prune_worktrees(cfg: Config) -> list[str]:
    """Drop registry entries whose git worktree no longer exists, and
    `git worktree prune` the repo."""
    root = cfg.project_root
    removed = []
    if root is not None and is_git_repo(root):
        _git(root, "worktree", "prune", check=False)
    for w in list_worktrees(cfg):
        if w["status"] == "stale":
            ProjectRegistry().remove_worktree(cfg.project_slug, w["branch"])
            removed.append(w["branch"])
    return removed
