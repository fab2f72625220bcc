"""Two-stage project image build.

Reference: internal/bundler (Dockerfile.base.tmpl + Dockerfile.harness-
image.tmpl generation; base-hash staleness basehash.go; egress floor
composition egress.go) + internal/docker/builder.go Build.

MI355X-first shape: no Dockerfiles and no registry — build scripts run in
overlay sandboxes over the hostfs base (the node's ROCm userland IS the
base image), producing:
  clawker-<project>:base      user setup + stacks + project packages/steps
  clawker-<project>:<harness> harness install + seeds + plan scripts +
                              managed prompt + egress floor (also tagged
                              :default)
"""
from __future__ import annotations

import base64
import hashlib
import os
import shlex

import yaml

from ..bundle import Harness, load_harness, load_stack
from ..bundle.loader import BuildStep, harness_egress_floor
from ..config import Config
from ..engine import Engine
from ..engine.build import ProgressFn, build_image
from ..engine.images import HOSTFS
from ..logger import get as get_logger
from ..storage import to_plain

log = get_logger("bundler")

AGENT_PROMPT = """\
# Agent briefing (managed by clawker)

You are running inside a clawker-amd sandbox on an AMD MI355X node.
- Your workspace is mounted at /workspace (or the path in $PWD).
- Egress is deny-by-default; allowed destinations are project policy.
- GPUs assigned to you: $CLAWKER_GPU (ROCm; use `rocm-smi` or torch).
- Do not attempt to escape the sandbox or disable the firewall.
"""


def _emit_file(path: str, content: str, mode: str = "644") -> str:
    b64 = base64.b64encode(content.encode()).decode()
    d = os.path.dirname(path)
    return (f"mkdir -p {shlex.quote(d)}\n"
            f"printf %s {shlex.quote(b64)} | base64 -d > {shlex.quote(path)}\n"
            f"chmod {mode} {shlex.quote(path)}\n")


def _steps_script(steps: list[BuildStep]) -> str:
    out = []
    for s in steps:
        if s.best_effort:
            out.append(f"({s.run}) || echo 'clawker: best-effort step failed: '{shlex.quote(s.run)}")
        else:
            out.append(s.run)
    return "\n".join(out)


class Builder:
    def __init__(self, cfg: Config, engine: Engine):
        self.cfg = cfg
        self.engine = engine

    # ---------------------------------------------------------------- base --
    def base_script(self, harness: Harness) -> str:
        proj = self.cfg.project.build
        uid = os.environ.get("SUDO_UID") or str(os.getuid() or 1000)
        gid = os.environ.get("SUDO_GID") or str(os.getgid() or 1000)
        user = harness.user or proj.user or "agent"
        lines = ["export DEBIAN_FRONTEND=noninteractive"]
        if user and user != "root":
            # host uid/gid so bind-mode workspace files keep ownership
            # (reference: Dockerfile.base.tmpl user setup with host UID/GID)
            lines.append(
                f"id -u {user} >/dev/null 2>&1 || "
                f"(groupadd -g {gid} {user} 2>/dev/null || true; "
                f"useradd -m -u {uid} -g {gid} -s /bin/bash {user} || "
                f"useradd -m -s /bin/bash {user})")
        stack_names = list(dict.fromkeys(list(harness.stacks) + list(proj.stacks)))
        pkgs = list(proj.packages)
        for sname in stack_names:
            st = load_stack(sname, self.cfg.project_root)
            pkgs += st.packages
            lines.append(_steps_script(st.steps))
        if pkgs:
            lines.append(
                "(apt-get update && apt-get install -y " + " ".join(map(shlex.quote, pkgs)) +
                ") || echo 'clawker: package install skipped (no network?)'")
        for step in proj.steps:
            lines.append(step)
        return "\n".join(l for l in lines if l)

    # ------------------------------------------------------------- harness --
    def harness_script(self, harness: Harness) -> str:
        lines = ["export DEBIAN_FRONTEND=noninteractive"]
        install_steps = harness.install
        if any("@VERSION@" in st.run for st in install_steps):
            from ..bundle.versions import resolve_version
            from ..config.config import load_settings
            try:
                settings = load_settings().get()
            except Exception:
                settings = None
            ver, src = resolve_version(harness.version, settings)
            lines.append(f"# harness version {ver} (source: {src})")
            install_steps = [
                BuildStep(run=st.run.replace("@VERSION@", ver),
                          best_effort=st.best_effort)
                for st in install_steps]
        lines.append(_steps_script(install_steps))
        # plan scripts consumed by controlplane/plans.py
        if harness.post_init:
            lines.append(_emit_file("/etc/clawker/post-init.sh", harness.post_init, "755"))
        if harness.pre_run:
            lines.append(_emit_file("/etc/clawker/pre-run.sh", harness.pre_run, "755"))
        for path, content in (harness.seeds or {}).items():
            lines.append(_emit_file(f"/etc/clawker/seeds{path}", str(content)))
        if harness.seeds:
            seed_apply = "#!/bin/sh\ncp -a /etc/clawker/seeds/. \"$HOME\"/ 2>/dev/null || true\n"
            lines.append(_emit_file("/etc/clawker/seed-apply.sh", seed_apply, "755"))
        # managed agent briefing (reference: clawker-agent-prompt.md baked at
        # the harness-declared path, bundler/dockerfile.go:57)
        prompt_path = harness.managed_prompt_path or self.cfg.project.agent.managed_prompt
        if prompt_path:
            lines.append(_emit_file(prompt_path, AGENT_PROMPT))
        # egress floor recorded in-image for the firewall to compose
        floor = harness_egress_floor(harness, self.cfg.project.security.egress)
        lines.append(_emit_file(
            "/etc/clawker/egress-floor.yaml",
            yaml.safe_dump({"harness": harness.name,
                            "rules": [to_plain(r) for r in floor]})))
        return "\n".join(l for l in lines if l)

    # ---------------------------------------------------------------- build --
    def build(self, harness_name: str = "", no_cache: bool = False,
              on_progress: ProgressFn | None = None) -> str:
        """Build base (if stale) + harness image; returns the image name."""
        harness = load_harness(
            harness_name or self.cfg.project.agent.harness, self.cfg.project_root)
        base_name = self.cfg.base_image_name()
        bscript = self.base_script(harness)
        bhash = hashlib.sha256(bscript.encode()).hexdigest()[:16]

        stale = no_cache or not self.engine.images.exists(base_name)
        if not stale:
            stale = self.engine.images.get(base_name).base_hash != bhash
        if stale:
            if on_progress:
                on_progress(f"building base image {base_name}")
            build_image(self.engine, base_name, HOSTFS, bscript,
                        base_hash=bhash, on_progress=on_progress)
        elif on_progress:
            on_progress(f"base image {base_name} up to date")

        hscript = self.harness_script(harness)
        img_name = self.cfg.image_name(harness.name)
        if on_progress:
            on_progress(f"building harness image {img_name}")
        env = dict(harness.env)
        for sname in harness.stacks:
            env.update(load_stack(sname, self.cfg.project_root).env)
        env.update(self.cfg.project.build.env)
        build_image(
            self.engine, img_name, base_name, hscript, env=env,
            user=harness.user or self.cfg.project.build.user,
            cmd=harness.cmd,
            labels={"dev.clawker.harness": harness.name},
            on_progress=on_progress)
        self.engine.images.tag(img_name, f"clawker-{self.cfg.project_slug}:default")
        log.info("build_done", image=img_name)
        return img_name
