"""Content-addressed overlay image store.

The reference builds Docker images (internal/docker/builder.go + bundler
Dockerfile generation). This node has no Docker and no network, so images
are *overlayfs layer stacks over the host filesystem* ("hostfs base"): the
ROCm dev userland already on the node is the base; a build runs its steps
in a throwaway sandbox whose upper dir becomes a new content-addressed
layer; instantiating a sandbox is just an overlay mount of
hostfs + image layers (ro) + a per-sandbox upper (rw). Zero pull, zero
copy on create — this is where the cold-start win comes from.

Store layout:
  images/layers/<id>/fs/      layer content
  images/meta/<name>.json     {"name","layers":[bottom..top],"env","user",
                               "cmd","workdir","labels","created","parent"}
"""
from __future__ import annotations

import hashlib
import io
import json
import os
import shutil
import tarfile
import time
import uuid
from dataclasses import dataclass, field
from pathlib import Path

from .. import consts
from ..errors import ConflictError, NotFoundError

HOSTFS = "hostfs"


def _enc(name: str) -> str:
    return name.replace("/", "%2F").replace(":", "%3A")


@dataclass
class ImageMeta:
    name: str
    layers: list[str] = field(default_factory=list)   # bottom..top layer ids
    env: dict = field(default_factory=dict)
    user: str = ""
    cmd: list = field(default_factory=list)
    workdir: str = ""
    labels: dict = field(default_factory=dict)
    created: float = 0.0
    parent: str = HOSTFS
    base_hash: str = ""     # staleness check input hash (reference: basehash.go)

    def to_dict(self) -> dict:
        return self.__dict__.copy()

    @classmethod
    def from_dict(cls, d: dict) -> "ImageMeta":
        known = {f for f in cls.__dataclass_fields__}  # type: ignore[attr-defined]
        return cls(**{k: v for k, v in d.items() if k in known})


class ImageStore:
    def __init__(self, root: Path | None = None):
        self.root = root or consts.image_store_dir()
        (self.root / "layers").mkdir(parents=True, exist_ok=True)
        (self.root / "meta").mkdir(parents=True, exist_ok=True)
        self._layerfs_checked = False

    def ensure_layer_filesystem(self) -> None:
        """Overlayfs forbids a lowerdir nested inside another lowerdir
        (mount-time ELOOP trap), and our bottom lower is the host "/" — so
        the layer store must live on its OWN superblock. If images/layers
        shares st_dev with /, mount a tmpfs over it (host-wide, survives
        the process; a production node should dedicate a real partition or
        bind a non-root filesystem at CLAWKER_IMAGE_DIR instead)."""
        if self._layerfs_checked:
            return
        self._layerfs_checked = True
        layers = self.root / "layers"
        try:
            if os.stat(layers).st_dev != os.stat("/").st_dev:
                return   # already separate (dedicated partition / prior mount)
            if os.geteuid() != 0:
                return
            import subprocess
            r = subprocess.run(
                ["mount", "-t", "tmpfs", "-o", "mode=700", "clawker-layers",
                 str(layers)], capture_output=True, text=True)
            if r.returncode != 0:
                from ..logger import get as _get
                _get("images").warn("layerfs_unavailable", err=r.stderr.strip())
        except OSError:
            pass

    # -- metadata ------------------------------------------------------------
    def _meta_path(self, name: str) -> Path:
        return self.root / "meta" / (_enc(name) + ".json")

    def exists(self, name: str) -> bool:
        return name == HOSTFS or self._meta_path(name).is_file()

    def get(self, name: str) -> ImageMeta:
        if name == HOSTFS:
            return ImageMeta(name=HOSTFS, created=0.0, parent="")
        p = self._meta_path(name)
        if not p.is_file():
            raise NotFoundError(f"image not found: {name}")
        return ImageMeta.from_dict(json.loads(p.read_text()))

    def put(self, meta: ImageMeta, overwrite: bool = True) -> None:
        meta.created = meta.created or time.time()
        p = self._meta_path(meta.name)
        if p.exists() and not overwrite:
            raise ConflictError(f"image exists: {meta.name}")
        p.write_text(json.dumps(meta.to_dict(), indent=1))

    def list(self) -> list[ImageMeta]:
        out = []
        for p in sorted((self.root / "meta").glob("*.json")):
            out.append(ImageMeta.from_dict(json.loads(p.read_text())))
        return out

    def remove(self, name: str, prune_layers: bool = True) -> None:
        meta = self.get(name)
        self._meta_path(name).unlink(missing_ok=True)
        if prune_layers:
            self.prune_layers()

    def tag(self, src: str, dst: str) -> None:
        meta = self.get(src)
        meta.name = dst
        self.put(meta)

    # -- layers ----------------------------------------------------------------
    def new_layer_dir(self) -> tuple[str, Path]:
        """Allocate a staging layer; caller fills fs/ then commit_layer()."""
        self.ensure_layer_filesystem()
        lid = "tmp-" + uuid.uuid4().hex[:12]
        d = self.root / "layers" / lid / "fs"
        d.mkdir(parents=True)
        # overlay workdir must share the upper's filesystem
        (self.root / "layers" / lid / "work").mkdir()
        return lid, d

    def commit_layer(self, tmp_id: str) -> str:
        """Rename a staging layer to its content id (cheap pseudo-hash:
        file list + sizes + mtimes digest — enough for staleness checks)."""
        src = self.root / "layers" / tmp_id
        shutil.rmtree(src / "work", ignore_errors=True)
        h = hashlib.sha256()
        fs = src / "fs"
        for p in sorted(fs.rglob("*")):
            st = p.lstat()
            h.update(str(p.relative_to(fs)).encode())
            h.update(f"{st.st_mode}:{st.st_size}".encode())
        lid = h.hexdigest()[:24]
        dst = self.root / "layers" / lid
        if dst.exists():
            shutil.rmtree(src)
        else:
            os.replace(src, dst)
        return lid

    def layer_path(self, lid: str) -> Path:
        p = self.root / "layers" / lid / "fs"
        if not p.is_dir():
            raise NotFoundError(f"layer missing: {lid}")
        return p

    def lowerdirs_for(self, name: str) -> list[str]:
        """Overlay lowerdirs, TOP-most first (ckrt joins with ':')."""
        meta = self.get(name)
        dirs: list[str] = []
        cur: ImageMeta | None = meta
        seen = set()
        while cur is not None:
            for lid in reversed(cur.layers):   # top layer of this image first
                dirs.append(str(self.layer_path(lid)))
            parent = cur.parent
            if not parent or parent in seen:
                break
            seen.add(parent)
            if parent == HOSTFS:
                dirs.append("/")
                break
            cur = self.get(parent)
        if not dirs:
            dirs.append("/")
        return dirs

    def prune_layers(self) -> int:
        """Remove layers referenced by no image. Staging layers (tmp-*) of
        a possibly in-flight build are only pruned once stale (>2h) — a
        concurrent build's upper dir must never vanish underneath it."""
        referenced: set[str] = set()
        for meta in self.list():
            referenced.update(meta.layers)
        removed = 0
        now = time.time()
        for d in (self.root / "layers").iterdir():
            if d.name in referenced:
                continue
            if d.name.startswith("tmp-"):
                try:
                    if now - d.stat().st_mtime < 2 * 3600:
                        continue
                except OSError:
                    continue
            shutil.rmtree(d, ignore_errors=True)
            removed += 1
        return removed

    # -------------------------------------------------------- save / load --
    def _image_chain(self, name: str) -> list[ImageMeta]:
        """The image plus its parents up to (not incl.) hostfs."""
        chain: list[ImageMeta] = []
        seen: set[str] = set()
        cur: ImageMeta | None = self.get(name)
        while cur is not None:
            chain.append(cur)
            parent = cur.parent
            if not parent or parent == HOSTFS or parent in seen:
                break
            seen.add(parent)
            cur = self.get(parent)
        return chain

    def save(self, name: str, out_path: Path) -> Path:
        """Export an image — manifest for it AND its parent chain, plus
        every referenced layer tree — as a tarball (docker-save analog).
        Useful because the layer store defaults to a tmpfs superblock
        (node reboot loses built images) and for moving images between
        nodes without rebuilding."""
        chain = self._image_chain(name)
        out_path = Path(out_path)
        out_path.parent.mkdir(parents=True, exist_ok=True)
        with tarfile.open(out_path, "w:gz") as tar:
            manifest = json.dumps(
                {"image": name, "images": [m.to_dict() for m in chain]},
                indent=1).encode()
            info = tarfile.TarInfo("manifest.json")
            info.size = len(manifest)
            info.mtime = int(time.time())
            tar.addfile(info, io.BytesIO(manifest))
            done: set[str] = set()
            for m in chain:
                for lid in m.layers:
                    if lid not in done:
                        done.add(lid)
                        tar.add(self.root / "layers" / lid,
                                arcname=f"layers/{lid}")
        return out_path

    def load(self, in_path: Path, rename: str = "") -> ImageMeta:
        """Import a tarball written by save(). Layers already present
        (same content id) are skipped; the image record is registered
        under its saved name (or `rename`)."""
        self.ensure_layer_filesystem()
        with tarfile.open(in_path) as tar:
            names = tar.getnames()
            if "manifest.json" not in names:
                raise ConflictError(f"{in_path}: not a clawker image tarball")
            # refuse traversal: every member stays under layers/ or is the
            # manifest (absolute paths and .. segments rejected)
            for m in tar.getmembers():
                n = m.name
                parts = Path(n).parts
                if (n != "manifest.json" and parts[:1] != ("layers",)) \
                        or ".." in parts or n.startswith("/"):
                    raise ConflictError(f"{in_path}: unsafe member {n!r}")
                # sym/hardlinks must stay inside the tree: an absolute or
                # ..-escaping linkname could alias host paths into the
                # store (write-through or read exposure)
                if m.issym() or m.islnk():
                    ln = m.linkname
                    if ln.startswith("/") or ".." in Path(ln).parts:
                        raise ConflictError(
                            f"{in_path}: unsafe link {n!r} -> {ln!r}")
            manifest = json.loads(tar.extractfile("manifest.json").read())
            if "images" in manifest:      # chain format
                metas = [ImageMeta.from_dict(d) for d in manifest["images"]]
                top = next(m for m in metas if m.name == manifest["image"])
            else:                         # single-image format
                metas = [ImageMeta.from_dict(manifest)]
                top = metas[0]
            for meta in metas:
                for lid in meta.layers:
                    dst = self.root / "layers" / lid
                    if dst.exists():
                        continue
                    members = [m for m in tar.getmembers()
                               if Path(m.name).parts[:2] == ("layers", lid)]
                    if not members:
                        raise ConflictError(
                            f"{in_path}: layer {lid} missing from tarball")
                    tar.extractall(self.root, members=members,
                                   numeric_owner=True)
        if rename:
            top.name = rename
            # parents keep their names: the renamed top still chains to them
        for meta in metas[::-1]:          # parents first
            if meta is not top and self.exists(meta.name):
                continue                  # never clobber an existing parent
            meta.created = time.time()
            self.put(meta)
        return top
