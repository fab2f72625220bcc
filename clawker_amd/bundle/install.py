"""Bundle git-install pipeline + GC (reference: internal/bundle
install.go fetchIntoCache — receipt-gated value-keyed cache, symlink
escape sanitization; manager.go GC against declaration roots).

Air-gapped nodes still benefit: `file://` and local-path git URLs work
offline, and a mirror URL works the moment one exists.
"""
from __future__ import annotations

import hashlib
import json
import shutil
import subprocess
import time
from pathlib import Path

import yaml

from .. import consts
from ..errors import ClawkerError
from ..logger import get as get_logger

log = get_logger("bundle")

MANIFESTS = ("harness.yaml", "stack.yaml", "manifest.yaml")


def cache_dir() -> Path:
    return consts.data_dir() / "bundle-cache"


def _cache_key(url: str, ref: str) -> str:
    return hashlib.sha256(f"{url}#{ref}".encode()).hexdigest()[:16]


def sanitize_symlinks(root: Path) -> int:
    """Remove symlinks that escape the bundle root (reference:
    install.go symlink-escape sanitization). In-tree relative links are
    kept; absolute or escaping links are dropped."""
    removed = 0
    for p in sorted(root.rglob("*"), reverse=True):
        if not p.is_symlink():
            continue
        try:
            target = p.resolve()
            if not str(target).startswith(str(root.resolve()) + "/") \
                    and target != root.resolve():
                p.unlink()
                removed += 1
        except OSError:
            try:
                p.unlink()
                removed += 1
            except OSError:
                pass
    return removed


def fetch_into_cache(url: str, ref: str = "") -> Path:
    """Clone (or reuse) a bundle repo in the value-keyed cache. A cache
    entry is valid only with a complete receipt (a crashed fetch is
    re-done, never half-trusted)."""
    key = _cache_key(url, ref)
    entry = cache_dir() / key
    receipt = entry / ".receipt.json"
    if receipt.is_file():
        try:
            rec = json.loads(receipt.read_text())
            if rec.get("url") == url and rec.get("ref", "") == ref:
                return entry
        except (ValueError, OSError):
            pass
    if entry.exists():
        shutil.rmtree(entry)
    entry.parent.mkdir(parents=True, exist_ok=True)
    tmp = entry.with_suffix(".fetch")
    if tmp.exists():
        shutil.rmtree(tmp)
    cmd = ["git", "clone", "--depth", "1"]
    if ref:
        cmd += ["--branch", ref]
    cmd += [url, str(tmp)]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=300)
    if r.returncode != 0:
        raise ClawkerError(f"bundle fetch failed: {r.stderr.strip()[-400:]}")
    commit = subprocess.run(["git", "-C", str(tmp), "rev-parse", "HEAD"],
                            capture_output=True, text=True).stdout.strip()
    shutil.rmtree(tmp / ".git", ignore_errors=True)
    dropped = sanitize_symlinks(tmp)
    tmp.replace(entry)
    (entry / ".receipt.json").write_text(json.dumps(
        {"url": url, "ref": ref, "commit": commit,
         "fetched": time.time(), "symlinks_dropped": dropped}))
    log.info("bundle_fetched", url=url, ref=ref, commit=commit[:12],
             symlinks_dropped=dropped)
    return entry


def _detect(root: Path) -> tuple[str, str, Path]:
    """Find (kind, name, bundle_dir) inside a fetched repo: the manifest
    may be at the top level or one directory deep."""
    candidates = [root] + sorted(
        p for p in root.iterdir() if p.is_dir() and not p.name.startswith("."))
    for d in candidates:
        for mf in MANIFESTS:
            f = d / mf
            if f.is_file():
                kind = ("stacks" if mf == "stack.yaml" else "harnesses")
                if mf == "manifest.yaml":
                    try:
                        doc = yaml.safe_load(f.read_text()) or {}
                        kind = ("stacks" if doc.get("kind") == "stack"
                                else "harnesses")
                    except yaml.YAMLError:
                        pass
                try:
                    doc = yaml.safe_load(f.read_text()) or {}
                    name = doc.get("name") or d.name
                except yaml.YAMLError:
                    name = d.name
                return kind, str(name), d
    raise ClawkerError(
        f"no bundle manifest ({'/'.join(MANIFESTS)}) found in {root}")


def install_from_git(url: str, ref: str = "", name: str = "") -> tuple[str, str]:
    """Fetch + install into the user tier. Returns (kind, name)."""
    if "#" in url and not ref:
        url, _, ref = url.partition("#")
    entry = fetch_into_cache(url, ref)
    kind, detected, bundle_dir = _detect(entry)
    name = name or detected
    dst = consts.config_dir() / kind / name
    if dst.exists():
        shutil.rmtree(dst)
    dst.parent.mkdir(parents=True, exist_ok=True)
    shutil.copytree(bundle_dir, dst, symlinks=True,
                    ignore=shutil.ignore_patterns(".receipt.json"))
    (dst / ".installed").write_text(json.dumps(
        {"at": time.time(), "source": url, "ref": ref}))
    return kind, name


def declared_components() -> tuple[set, set]:
    """Harness/stack names declared by registered projects + global
    settings — the GC roots (reference: GC against declaration roots)."""
    harnesses: set[str] = set()
    stacks: set[str] = set()
    from ..project.registry import ProjectRegistry
    try:
        entries = ProjectRegistry().list_projects()
    except Exception:
        entries = []
    for e in entries:
        root = Path(getattr(e, "root", "") or "")
        for cand in (root / ".clawker.yaml",
                     root / consts.PROJECT_DIR_NAME / "clawker.yaml"):
            try:
                doc = yaml.safe_load(cand.read_text()) or {}
            except (OSError, yaml.YAMLError):
                continue
            agent = doc.get("agent") or {}
            if agent.get("harness"):
                harnesses.add(str(agent["harness"]))
            for s in (doc.get("build") or {}).get("stacks", []) or []:
                stacks.add(str(s))
    return harnesses, stacks


def gc(dry_run: bool = False) -> dict:
    """Remove user-tier installed bundles and cache entries that no
    registered project declares. Loose (hand-placed, no .installed
    receipt) user bundles are never touched."""
    harnesses, stacks = declared_components()
    removed = {"harnesses": [], "stacks": [], "cache": []}
    for kind, declared in (("harnesses", harnesses), ("stacks", stacks)):
        d = consts.config_dir() / kind
        if not d.is_dir():
            continue
        for p in d.iterdir():
            if not p.is_dir() or not (p / ".installed").is_file():
                continue
            if p.name not in declared:
                removed[kind].append(p.name)
                if not dry_run:
                    shutil.rmtree(p)
    # cache entries are value-keyed; an entry whose install target is
    # gone (or was GC'd above) has no referent
    installed_sources = set()
    for kind in ("harnesses", "stacks"):
        d = consts.config_dir() / kind
        if d.is_dir():
            for p in d.iterdir():
                f = p / ".installed"
                if f.is_file():
                    try:
                        installed_sources.add(
                            json.loads(f.read_text()).get("source", ""))
                    except (ValueError, OSError):
                        pass
    if cache_dir().is_dir():
        for entry in cache_dir().iterdir():
            rec = entry / ".receipt.json"
            if not rec.is_file():
                removed["cache"].append(entry.name)
                if not dry_run:
                    shutil.rmtree(entry, ignore_errors=True)
                continue
            try:
                url = json.loads(rec.read_text()).get("url", "")
            except (ValueError, OSError):
                url = ""
            if url not in installed_sources:
                removed["cache"].append(entry.name)
                if not dry_run:
                    shutil.rmtree(entry, ignore_errors=True)
    return removed
