"""Generic layered-YAML store engine.

Reference: internal/storage (store.go:83 New, merge.go, write.go,
discover.go) — a generic ``Store[T]``: layer discovery (walk-up with
``.clawker/`` dir-vs-dotfile dual placement), per-layer migrations, N-way
merge with ``merge:"union"|"overwrite"`` struct tags and provenance (which
layer won each field), atomic temp+rename persistence under an flock.

This rebuild keeps the same semantics on Python dataclasses: merge tags come
from ``dataclasses.field(metadata={"merge": "union"})``; provenance is a
dotted-path -> layer-name map that drives auto-routed writes. Comments
survive rewrites via verified text surgery (yamledit.py) with a plain-dump
fallback when surgery is unsafe.
"""
from __future__ import annotations

import dataclasses
import fcntl
import os
import tempfile
import typing as t
from pathlib import Path

import yaml

from ..errors import ClawkerError

T = t.TypeVar("T")

DEFAULTS_LAYER = "defaults"


class StoreError(ClawkerError):
    pass


@dataclasses.dataclass
class Layer:
    """One YAML document in the stack. Later layers win (higher priority)."""

    name: str
    path: Path | None          # None => virtual (defaults) layer
    writable: bool = True
    data: dict = dataclasses.field(default_factory=dict)
    exists: bool = False


@dataclasses.dataclass
class MergeResult:
    merged: dict
    # dotted field path -> winning layer name (leaf scalars and whole lists)
    provenance: dict[str, str] = dataclasses.field(default_factory=dict)


# ------------------------------------------------------------- dataclass ----

def _is_dc(tp: t.Any) -> bool:
    return dataclasses.is_dataclass(tp) and isinstance(tp, type)


def _strip_optional(tp: t.Any) -> t.Any:
    origin = t.get_origin(tp)
    if origin is t.Union:
        args = [a for a in t.get_args(tp) if a is not type(None)]
        if len(args) == 1:
            return args[0]
    return tp


def dataclass_defaults(tp: type) -> dict:
    """Materialize the virtual defaults layer from field defaults
    (reference: defaults from struct tags as a virtual lowest layer)."""
    out: dict = {}
    for f in dataclasses.fields(tp):
        ft = _strip_optional(f.type if not isinstance(f.type, str) else _resolve_hint(tp, f.name))
        if _is_dc(ft):
            out[f.name] = dataclass_defaults(ft)
        elif f.default is not dataclasses.MISSING:
            out[f.name] = _plain(f.default)
        elif f.default_factory is not dataclasses.MISSING:  # type: ignore[misc]
            out[f.name] = _plain(f.default_factory())  # type: ignore[misc]
    return out


def _resolve_hint(tp: type, field_name: str) -> t.Any:
    hints = _hints(tp)
    return hints.get(field_name, t.Any)


def _plain(v: t.Any) -> t.Any:
    if dataclasses.is_dataclass(v) and not isinstance(v, type):
        return {f.name: _plain(getattr(v, f.name)) for f in dataclasses.fields(v)}
    if isinstance(v, dict):
        return {k: _plain(x) for k, x in v.items()}
    if isinstance(v, (list, tuple)):
        return [_plain(x) for x in v]
    return v


def to_plain(v: t.Any) -> t.Any:
    """Public: dataclass instance -> plain dict tree (for YAML dump)."""
    return _plain(v)


_hints_cache: dict[type, dict] = {}


def _hints(tp: type) -> dict:
    """get_type_hints is surprisingly expensive (string annotations are
    compile()d per call — ~16 ms per sandbox create before caching);
    schema classes never change at runtime."""
    h = _hints_cache.get(tp)
    if h is None:
        h = t.get_type_hints(tp)
        _hints_cache[tp] = h
    return h


def materialize(tp: type[T], data: dict) -> T:
    """Construct a (nested) dataclass instance from a merged dict, applying
    field defaults for anything missing and ignoring unknown keys."""
    kwargs: dict[str, t.Any] = {}
    hints = _hints(tp)
    for f in dataclasses.fields(tp):  # type: ignore[arg-type]
        ft = _strip_optional(hints.get(f.name, t.Any))
        if f.name in data and data[f.name] is not None:
            v = data[f.name]
            if _is_dc(ft):
                if isinstance(v, dict):
                    kwargs[f.name] = materialize(ft, v)
                else:
                    raise StoreError(f"field {f.name}: expected mapping, got {type(v).__name__}")
            elif t.get_origin(ft) is list and v is not None:
                (item_tp,) = t.get_args(ft) or (t.Any,)
                item_tp = _strip_optional(item_tp)
                if _is_dc(item_tp):
                    kwargs[f.name] = [materialize(item_tp, x) if isinstance(x, dict) else x for x in v]
                else:
                    kwargs[f.name] = list(v)
            elif t.get_origin(ft) is dict and isinstance(v, dict):
                kwargs[f.name] = dict(v)
            else:
                kwargs[f.name] = v
        elif f.default is dataclasses.MISSING and f.default_factory is dataclasses.MISSING:  # type: ignore[misc]
            if _is_dc(ft):
                kwargs[f.name] = materialize(ft, {})
            else:
                raise StoreError(f"missing required field: {f.name}")
    return tp(**kwargs)  # type: ignore[return-value]


def merge_tags(tp: type) -> dict[str, str]:
    """Dotted path -> merge strategy ("union"|"overwrite") from field metadata."""
    tags: dict[str, str] = {}

    def walk(dc: type, prefix: str) -> None:
        hints = _hints(dc)
        for f in dataclasses.fields(dc):
            path = f"{prefix}{f.name}"
            tag = f.metadata.get("merge")
            if tag:
                tags[path] = tag
            ft = _strip_optional(hints.get(f.name, t.Any))
            if _is_dc(ft):
                walk(ft, path + ".")

    if _is_dc(tp):
        walk(tp, "")
    return tags


# ----------------------------------------------------------------- merge ----

def merge_layers(layers: list[Layer], tags: dict[str, str] | None = None) -> MergeResult:
    """N-way merge, later layers win. Dicts deep-merge; lists overwrite unless
    their dotted path is tagged ``union`` (dedupe-preserving append);
    scalars overwrite. Provenance records the winning layer per leaf."""
    tags = tags or {}
    merged: dict = {}
    prov: dict[str, str] = {}

    def merge_into(dst: dict, src: dict, layer: str, prefix: str) -> None:
        for k, v in src.items():
            path = f"{prefix}{k}" if not prefix else f"{prefix}.{k}"
            if isinstance(v, dict) and isinstance(dst.get(k), dict):
                merge_into(dst[k], v, layer, path)
            elif isinstance(v, dict):
                dst[k] = {}
                merge_into(dst[k], v, layer, path)
            elif isinstance(v, list) and tags.get(path) == "union" and isinstance(dst.get(k), list):
                for item in v:
                    if item not in dst[k]:
                        dst[k].append(item)
                prov[path] = layer  # union: last contributor recorded
            else:
                dst[k] = v.copy() if isinstance(v, list) else v
                prov[path] = layer

    for layer in layers:
        if layer.data:
            merge_into(merged, layer.data, layer.name, "")
    return MergeResult(merged=merged, provenance=prov)


# ------------------------------------------------------------- discovery ----

def discover_project_layers(start: Path, stop_at: Path | None = None) -> list[Path]:
    """Walk up from ``start`` collecting project config files, nearest last
    (= highest priority). At each dir, ``.clawker/clawker.yaml`` (dir form)
    wins over ``.clawker.yaml`` (dotfile form); a ``clawker.local.yaml``
    sibling layers above its ``clawker.yaml``.
    Reference: internal/storage/discover.go walk-up + dual placement.
    """
    from .. import consts

    groups: list[list[Path]] = []
    cur = start.resolve()
    stop = stop_at.resolve() if stop_at else None
    while True:
        dir_form = cur / consts.PROJECT_DIR_NAME / consts.PROJECT_CONFIG_BASENAME
        dot_form = cur / consts.PROJECT_FILE_NAME
        local_dir = cur / consts.PROJECT_DIR_NAME / consts.PROJECT_LOCAL_BASENAME
        local_dot = cur / (consts.PROJECT_LOCAL_BASENAME.replace("clawker", ".clawker", 1))
        group: list[Path] = []
        if dir_form.is_file():
            group.append(dir_form)
            if local_dir.is_file():
                group.append(local_dir)
        elif dot_form.is_file():
            group.append(dot_form)
            if local_dot.is_file():
                group.append(local_dot)
        if group:
            groups.append(group)
        if stop is not None and cur == stop:
            break
        if cur.parent == cur:
            break
        cur = cur.parent
    # farthest-from-start dir first so nearest (most specific) ends last /
    # wins; within a dir, the local layer stays above its clawker.yaml
    out: list[Path] = []
    for group in reversed(groups):
        out.extend(group)
    return out


# ------------------------------------------------------------------ store ---

Migration = t.Callable[[dict], dict]


class Store(t.Generic[T]):
    """Layered store over a dataclass schema.

    Layers are ordered lowest→highest priority; a virtual ``defaults`` layer
    (from dataclass field defaults) always sits at the bottom.
    """

    def __init__(
        self,
        schema: type[T],
        layers: list[Layer],
        migrations: list[tuple[int, Migration]] | None = None,
        defaults_layer: bool = True,
    ):
        self.schema = schema
        self._tags = merge_tags(schema)
        self._migrations = sorted(migrations or [], key=lambda m: m[0])
        self.layers: list[Layer] = []
        if defaults_layer:
            self.layers.append(
                Layer(name=DEFAULTS_LAYER, path=None, writable=False,
                      data=dataclass_defaults(schema), exists=True))
        self.layers.extend(layers)
        self._result: MergeResult | None = None
        self.reload()

    # -- construction helpers ------------------------------------------------
    @classmethod
    def from_string(cls, schema: type[T], text: str, name: str = "test") -> "Store[T]":
        """In-memory single-layer store (reference: storage.NewFromString)."""
        data = yaml.safe_load(text) or {}
        if not isinstance(data, dict):
            raise StoreError("top-level YAML must be a mapping")
        return cls(schema, [Layer(name=name, path=None, writable=True, data=data, exists=True)])

    # -- IO ------------------------------------------------------------------
    def reload(self) -> None:
        for layer in self.layers:
            if layer.path is None:
                continue
            layer.data, layer.exists = self._read_layer(layer.path)
            if layer.exists and self._migrations:
                layer.data, changed = self._apply_migrations(layer.data)
                if changed and layer.writable:
                    self._write_file(layer.path, layer.data)
        self._result = merge_layers(self.layers, self._tags)

    def _read_layer(self, path: Path) -> tuple[dict, bool]:
        try:
            text = path.read_text()
        except FileNotFoundError:
            return {}, False
        data = yaml.safe_load(text) or {}
        if not isinstance(data, dict):
            raise StoreError(f"{path}: top-level YAML must be a mapping")
        return data, True

    def _apply_migrations(self, data: dict) -> tuple[dict, bool]:
        version = int(data.get("version", 0) or 0)
        changed = False
        for v, fn in self._migrations:
            if version < v:
                data = fn(data)
                data["version"] = v
                version = v
                changed = True
        return data, changed

    # -- reads ---------------------------------------------------------------
    @property
    def merged(self) -> dict:
        assert self._result is not None
        return self._result.merged

    def get(self) -> T:
        return materialize(self.schema, self.merged)

    def get_path(self, dotted: str, default: t.Any = None) -> t.Any:
        cur: t.Any = self.merged
        for part in dotted.split("."):
            if not isinstance(cur, dict) or part not in cur:
                return default
            cur = cur[part]
        return cur

    def provenance(self, dotted: str) -> str | None:
        assert self._result is not None
        p = self._result.provenance
        if dotted in p:
            return p[dotted]
        # containers: highest-priority contributor among children
        best: str | None = None
        best_idx = -1
        names = [l.name for l in self.layers]
        for path, layer in p.items():
            if path.startswith(dotted + "."):
                idx = names.index(layer)
                if idx > best_idx:
                    best, best_idx = layer, idx
        return best

    # -- writes --------------------------------------------------------------
    def _layer(self, name: str) -> Layer:
        for l in self.layers:
            if l.name == name:
                return l
        raise StoreError(f"unknown layer: {name}")

    def _route_layer(self, dotted: str) -> Layer:
        """Auto-route a write: the layer that currently owns the field if
        writable, else the highest-priority writable layer
        (reference: merge provenance drives auto-routed writes)."""
        owner = self.provenance(dotted)
        if owner and owner != DEFAULTS_LAYER:
            l = self._layer(owner)
            if l.writable:
                return l
        for l in reversed(self.layers):
            if l.writable:
                return l
        raise StoreError("no writable layer")

    def set(self, dotted: str, value: t.Any, layer: str | None = None) -> str:
        """Set a field; returns the layer written."""
        target = self._layer(layer) if layer else self._route_layer(dotted)
        cur = target.data
        parts = dotted.split(".")
        for part in parts[:-1]:
            nxt = cur.get(part)
            if not isinstance(nxt, dict):
                nxt = {}
                cur[part] = nxt
            cur = nxt
        cur[parts[-1]] = _plain(value)
        self._result = merge_layers(self.layers, self._tags)
        return target.name

    def remove(self, dotted: str, layer: str | None = None) -> bool:
        targets = [self._layer(layer)] if layer else [l for l in self.layers if l.writable]
        removed = False
        for target in targets:
            cur: t.Any = target.data
            parts = dotted.split(".")
            for part in parts[:-1]:
                if not isinstance(cur, dict) or part not in cur:
                    cur = None
                    break
                cur = cur[part]
            if isinstance(cur, dict) and parts[-1] in cur:
                del cur[parts[-1]]
                removed = True
        if removed:
            self._result = merge_layers(self.layers, self._tags)
        return removed

    def write(self) -> None:
        """Persist every writable file-backed layer atomically under flock."""
        for layer in self.layers:
            if layer.writable and layer.path is not None:
                self._write_file(layer.path, layer.data)
                layer.exists = True

    def write_layer(self, name: str) -> None:
        layer = self._layer(name)
        if layer.path is None:
            raise StoreError(f"layer {name} is virtual")
        self._write_file(layer.path, layer.data)
        layer.exists = True

    @staticmethod
    def _write_file(path: Path, data: dict) -> None:
        path.parent.mkdir(parents=True, exist_ok=True)
        lock = path.with_suffix(path.suffix + ".lock")
        with open(lock, "w") as lf:
            fcntl.flock(lf, fcntl.LOCK_EX)
            try:
                # comment-preserving surgery when the file exists
                # (reference: write.go yaml.Node round trip); verified by
                # re-parse, falls back to a plain dump when unsafe
                surgical: str | None = None
                try:
                    if path.is_file():
                        from .yamledit import update_yaml_text
                        surgical = update_yaml_text(path.read_text(), data)
                except OSError:
                    surgical = None
                fd, tmp = tempfile.mkstemp(dir=str(path.parent), prefix="." + path.name)
                try:
                    with os.fdopen(fd, "w") as f:
                        if surgical is not None:
                            f.write(surgical)
                        else:
                            yaml.safe_dump(data, f, sort_keys=False,
                                           default_flow_style=False)
                    os.replace(tmp, path)
                except BaseException:
                    try:
                        os.unlink(tmp)
                    except OSError:
                        pass
                    raise
            finally:
                fcntl.flock(lf, fcntl.LOCK_UN)
