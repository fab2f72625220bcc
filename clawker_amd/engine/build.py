"""Image build primitive: run a build script in a throwaway sandbox whose
overlay upper becomes a new content-addressed image layer.

Reference analog: internal/docker/builder.go Build driving dockerd builds;
here the engine is its own builder (no daemon): layer = the COW residue of
the build sandbox. Requires the ns backend (no overlay on proc hosts)."""
from __future__ import annotations

import shutil
import subprocess
import time
import uuid
from pathlib import Path
from typing import Callable

from .. import consts
from ..errors import EngineError
from ..logger import get as get_logger
from .engine import Engine
from .images import ImageMeta
from .spec import SandboxSpec

log = get_logger("build")

ProgressFn = Callable[[str], None]


def build_layer(engine: Engine, base_image: str, script: str,
                env: dict | None = None, network: bool = True,
                on_progress: ProgressFn | None = None,
                timeout_s: float = 900.0) -> str:
    """Run `script` (sh -e) over base_image; returns the committed layer id.
    Build output streams through on_progress."""
    if engine.backend != "ns":
        raise EngineError(
            "build", "image builds need the ns isolation backend "
            "(overlayfs); this host cannot create mount namespaces")
    tmp_id, layer_fs = engine.images.new_layer_dir()
    name = f"{consts.SANDBOX_NAME_PREFIX}build.{uuid.uuid4().hex[:8]}"
    spec = SandboxSpec(
        name=name, hostname="clawker-build", netns=not network,
        autostart=True, cmd=["/bin/sh", "-ec", script],
        env=dict(env or {}), user="", workdir="/",
        labels={consts.MANAGED_LABEL: "true", "dev.clawker.build": "true"},
    )
    spec.upper = str(layer_fs)
    spec.work = str(layer_fs.parent / "work")
    engine.create(spec, image=base_image)
    try:
        engine.start(name)
        # stream console as progress
        last = 0
        deadline = time.monotonic() + timeout_s
        while True:
            data = engine.logs(name)
            if on_progress and len(data) > last:
                for line in data[last:].decode(errors="replace").splitlines():
                    on_progress(line)
                last = len(data)
            info = engine.inspect(name)
            if info.state != "running":
                code = info.exit_code
                if code is None:
                    # the shim writes exit.json right AFTER the child is
                    # reaped: a poll can land in that window. wait() holds
                    # through the grace loop for the definitive code.
                    try:
                        code = engine.wait(name, timeout_s=5)
                    except EngineError:
                        code = -1
                break
            if time.monotonic() > deadline:
                engine.stop(name, timeout_s=3)
                raise EngineError("build", f"build timed out after {timeout_s}s")
            time.sleep(0.05)
        if code != 0:
            tail = engine.logs(name)[-1200:].decode(errors="replace")
            raise EngineError("build", f"build script failed ({code}):\n{tail}")
    finally:
        try:
            engine.remove(name, force=True)
        except Exception:
            pass
    # scrub build-sandbox runtime residue from the layer
    for junk in ("run/clawker", ".oldroot"):
        p = layer_fs / junk
        if p.exists():
            import shutil
            shutil.rmtree(p, ignore_errors=True)
    lid = engine.images.commit_layer(tmp_id)
    log.info("layer_built", layer=lid, base=base_image)
    return lid


def build_image(engine: Engine, name: str, base_image: str, script: str,
                env: dict | None = None, user: str = "", cmd: list | None = None,
                workdir: str = "", labels: dict | None = None,
                base_hash: str = "", network: bool = True,
                on_progress: ProgressFn | None = None) -> ImageMeta:
    lid = build_layer(engine, base_image, script, env=env, network=network,
                      on_progress=on_progress)
    meta = ImageMeta(
        name=name, layers=[lid], env=dict(env or {}), user=user,
        cmd=list(cmd or []), workdir=workdir, labels=dict(labels or {}),
        parent=base_image, base_hash=base_hash)
    engine.images.put(meta)
    log.info("image_built", image=name, layer=lid)
    return meta


def commit_sandbox(engine: Engine, name: str, image_name: str,
                   message: str = "") -> ImageMeta:
    """docker-commit analog: snapshot a sandbox's copy-on-write upper as
    a new content-addressed layer stacked on its image. Overlay artifacts
    (whiteout device nodes, opaque-dir xattrs) are preserved via cp -a so
    deletions carry into the committed image; runtime residue (rundir
    mountpoint, identity files rewritten per-create anyway) is scrubbed."""
    info = engine.inspect(name)
    upper = Path(info.statedir) / "upper"
    if not upper.is_dir():
        raise EngineError("commit", f"no writable layer for {name}")
    # freeze a running workload while the layer is copied so the snapshot
    # is point-in-time consistent (docker commit --pause semantics)
    froze = False
    if info.state == "running":
        try:
            engine.pause(name)
            froze = True
        except Exception:
            pass
    tmp_id, layer_fs = engine.images.new_layer_dir()
    try:
        r = subprocess.run(
            ["cp", "-a", "--reflink=auto", f"{upper}/.", str(layer_fs)],
            capture_output=True, text=True)
    finally:
        if froze:
            engine.unpause(name)
    if r.returncode != 0:
        shutil.rmtree(layer_fs.parent, ignore_errors=True)
        raise EngineError("commit", f"layer copy failed: {r.stderr.strip()}")
    for junk in ("run/clawker", ".oldroot", "etc/hostname", "etc/hosts",
                 "etc/resolv.conf"):
        p = layer_fs / junk
        if p.is_dir() and not p.is_symlink():
            shutil.rmtree(p, ignore_errors=True)
        else:
            p.unlink(missing_ok=True)
    lid = engine.images.commit_layer(tmp_id)
    base = engine.images.get(info.image)
    # layers holds ONLY this image's own layer: lowerdirs_for composes the
    # full stack by walking the parent chain (duplicating base layers here
    # would mount the same lowerdir twice -> overlayfs ELOOP)
    meta = ImageMeta(
        name=image_name, layers=[lid], env=dict(base.env),
        user=base.user, cmd=list(base.cmd), workdir=base.workdir,
        labels={**base.labels, "dev.clawker.commit.from": name,
                **({"dev.clawker.commit.message": message} if message else {})},
        parent=info.image)
    engine.images.put(meta)
    log.info("sandbox_committed", sandbox=name, image=image_name, layer=lid)
    return meta
