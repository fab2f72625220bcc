"""Bundle git-install pipeline + GC + version resolvers (VERDICT r01
missing #5/#6; reference: internal/bundle install.go + manager.go GC,
internal/bundler/versions.go)."""
import json
import os
import subprocess
from pathlib import Path

import pytest


def _git(cwd, *args):
    r = subprocess.run(["git", *args], cwd=cwd, capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    return r.stdout


@pytest.fixture
def bundle_repo(tmp_path):
    """A local git repo holding a harness bundle (file:// works
    air-gapped)."""
    repo = tmp_path / "hx-repo"
    repo.mkdir()
    (repo / "harness.yaml").write_text(
        "name: hx\ncmd: [hx-agent]\nuser: agent\n")
    (repo / "seed.txt").write_text("seed!\n")
    # an escaping symlink that must be stripped
    os.symlink("/etc/passwd", repo / "evil-link")
    os.symlink("seed.txt", repo / "ok-link")
    _git(repo, "init", "-q")
    _git(repo, "config", "user.email", "t@t")
    _git(repo, "config", "user.name", "t")
    _git(repo, "add", "-A")
    _git(repo, "commit", "-qm", "bundle")
    return repo


def test_git_install_and_resolution(isolated_env, bundle_repo):
    from clawker_amd import consts
    from clawker_amd.bundle import load_harness
    from clawker_amd.bundle.install import install_from_git
    kind, name = install_from_git(f"file://{bundle_repo}")
    assert (kind, name) == ("harnesses", "hx")
    dst = consts.config_dir() / "harnesses" / "hx"
    assert (dst / "harness.yaml").is_file()
    assert not (dst / "evil-link").exists()       # escaping symlink dropped
    assert (dst / "ok-link").is_symlink()         # in-tree symlink kept
    h = load_harness("hx")
    assert h.cmd == ["hx-agent"]
    # receipt-gated cache: a second install reuses the entry
    from clawker_amd.bundle.install import cache_dir, fetch_into_cache
    entry = fetch_into_cache(f"file://{bundle_repo}")
    rec = json.loads((entry / ".receipt.json").read_text())
    assert rec["url"].startswith("file://")
    assert rec["commit"]
    assert rec["symlinks_dropped"] == 1


def test_gc_keeps_declared_removes_orphans(isolated_env, bundle_repo, tmp_path):
    from clawker_amd import consts
    from clawker_amd.bundle.install import gc, install_from_git
    from clawker_amd.project.registry import ProjectRegistry
    install_from_git(f"file://{bundle_repo}")                # name: hx
    install_from_git(f"file://{bundle_repo}", name="orphan") # undeclared
    # a project declaring harness hx
    proj = tmp_path / "gcproj"
    proj.mkdir()
    (proj / ".clawker.yaml").write_text(
        "project: gcp\nagent:\n  harness: hx\n")
    ProjectRegistry().register("gcp", proj)
    removed = gc(dry_run=True)
    assert "orphan" in removed["harnesses"]
    assert "hx" not in removed["harnesses"]
    removed = gc()
    assert not (consts.config_dir() / "harnesses" / "orphan").exists()
    assert (consts.config_dir() / "harnesses" / "hx").exists()
    # hand-placed (loose) bundles are never collected
    loose = consts.config_dir() / "harnesses" / "loose"
    loose.mkdir(parents=True)
    (loose / "harness.yaml").write_text("name: loose\ncmd: [x]\n")
    gc()
    assert loose.exists()


def test_version_resolver_pin_fallback(isolated_env):
    from clawker_amd.bundle.versions import VersionSpec, resolve_version
    v, src = resolve_version(VersionSpec(kind="npm", package="foo",
                                         pin="2.3.4"), settings=None)
    assert (v, src) == ("2.3.4", "pin")     # air-gapped: pin wins
    v, src = resolve_version(VersionSpec(pin="9.9.9"))
    assert (v, src) == ("9.9.9", "pin")


def test_version_resolver_registry_and_cache(isolated_env):
    import http.server
    import threading
    from types import SimpleNamespace
    from clawker_amd.bundle.versions import VersionSpec, resolve_version

    class H(http.server.BaseHTTPRequestHandler):
        calls = []

        def do_GET(self):
            H.calls.append(self.path)
            if self.path.startswith("/repos/"):
                body = json.dumps({"tag_name": "rust-v0.55.0"}).encode()
            else:
                body = json.dumps({"dist-tags": {"latest": "7.8.9"}}).encode()
            self.send_response(200)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, *a):
            pass

    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), H)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    base = f"http://127.0.0.1:{srv.server_address[1]}"
    settings = SimpleNamespace(bundler=SimpleNamespace(
        npm_registry=base, github_api=base))
    v, src = resolve_version(VersionSpec(kind="npm", package="pkg",
                                         pin="0.0.1"), settings)
    assert (v, src) == ("7.8.9", "registry")
    v2, src2 = resolve_version(VersionSpec(kind="npm", package="pkg",
                                           pin="0.0.1"), settings)
    assert (v2, src2) == ("7.8.9", "cache")   # TTL cache hit, no 2nd call
    v3, src3 = resolve_version(VersionSpec(
        kind="github-release", package="openai/codex",
        tag_prefix="rust-v", pin="0.1.0"), settings)
    assert (v3, src3) == ("0.55.0", "registry")   # prefix stripped
    srv.shutdown()


def test_builder_substitutes_version(isolated_env, tmp_path):
    """@VERSION@ in install steps expands to the resolved version."""
    from clawker_amd import consts
    hx = consts.config_dir() / "harnesses" / "vx"
    hx.mkdir(parents=True)
    (hx / "harness.yaml").write_text(
        "name: vx\ncmd: [vx]\n"
        "version: {kind: npm, package: vxpkg, pin: '3.1.4'}\n"
        "install:\n  - run: npm install -g vxpkg@@VERSION@\n")
    ws = tmp_path / "vproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text(
        "project: vproj\nagent:\n  harness: vx\n")
    from clawker_amd.bundle import load_harness
    from clawker_amd.bundler.builder import Builder
    from clawker_amd.config import load_config
    from clawker_amd.engine import Engine
    cfg = load_config(ws)
    text = Builder(cfg, Engine()).harness_script(load_harness("vx"))
    assert "vxpkg@3.1.4" in text
    assert "@VERSION@" not in text
    assert "source: pin" in text
