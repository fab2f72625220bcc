"""Interactive field editor over a Store[T].

Reference: internal/storeui — a reflection-driven field browser/editor
with per-field save and layer targeting (2.5k LoC BubbleTea TUI). This is
the prompter-based reduction: walk the schema's leaf fields with current
values and provenance, pick one, enter a new value (YAML-parsed), save
with the store's auto-routing.
"""
from __future__ import annotations

import dataclasses
import typing as t

import yaml

from .iostreams import IOStreams
from .prompter import Prompter
from .storage import Store


def leaf_fields(schema: type, prefix: str = "") -> list[str]:
    out: list[str] = []
    hints = t.get_type_hints(schema)
    for f in dataclasses.fields(schema):
        ft = hints.get(f.name, t.Any)
        origin = t.get_origin(ft)
        if origin is t.Union:
            args = [a for a in t.get_args(ft) if a is not type(None)]
            if len(args) == 1:
                ft = args[0]
        if dataclasses.is_dataclass(ft) and isinstance(ft, type):
            out.extend(leaf_fields(ft, f"{prefix}{f.name}."))
        else:
            out.append(f"{prefix}{f.name}")
    return out


def layer_values(store: Store, path: str) -> list[tuple[str, t.Any]]:
    """Each layer's raw value for a dotted path (layer browser pane —
    reference: storeui's per-layer inspection, edit.go:268)."""
    out: list[tuple[str, t.Any]] = []
    for layer in store.layers:
        cur: t.Any = layer.data
        for part in path.split("."):
            if not isinstance(cur, dict) or part not in cur:
                cur = _MISSING
                break
            cur = cur[part]
        if cur is not _MISSING:
            out.append((layer.name, cur))
    return out


_MISSING = object()


def edit_store(store: Store, io: IOStreams, max_rounds: int = 100) -> int:
    """Interactive loop; returns the number of fields changed.

    Commands at the selection prompt:
      <n>       edit field n (value prompt, then optional layer target)
      /text     filter the field list
      ?n        show every layer's value for field n
      (empty)   finish
    """
    p = Prompter(io)
    all_fields = leaf_fields(store.schema)
    fields = all_fields
    writable = [l.name for l in store.layers if l.writable]
    changed = 0
    for _ in range(max_rounds):
        io.print("")
        for i, path in enumerate(fields, 1):
            val = store.get_path(path)
            owner = store.provenance(path) or "defaults"
            io.print(f" {i:2}) {path} = {val!r}  [dim]({owner})[/dim]")
        sel = p.string(
            "field number to edit (/filter, ?n = layers, empty to finish)", "")
        sel = sel.strip()
        if not sel:
            break
        if sel.startswith("/"):
            pat = sel[1:].strip().lower()
            fields = ([f for f in all_fields if pat in f.lower()]
                      if pat else all_fields)
            if not fields:
                io.error(f"no fields match '{pat}'")
                fields = all_fields
            continue
        if sel.startswith("?"):
            try:
                path = fields[int(sel[1:]) - 1]
            except (ValueError, IndexError):
                io.error(f"invalid selection: {sel}")
                continue
            vals = layer_values(store, path)
            io.print(f"[bold]{path}[/bold] by layer "
                     f"(effective: {store.get_path(path)!r}):")
            for lname, v in vals or [("(none)", "unset everywhere")]:
                marker = "*" if store.provenance(path) == lname else " "
                io.print(f"  {marker} {lname}: {v!r}")
            continue
        try:
            idx = int(sel) - 1
            path = fields[idx]
        except (ValueError, IndexError):
            io.error(f"invalid selection: {sel}")
            continue
        raw = p.string(f"new value for {path} (YAML)", str(store.get_path(path)))
        try:
            value = yaml.safe_load(raw)
        except yaml.YAMLError as e:
            io.error(f"bad value: {e}")
            continue
        # layer targeting (reference: per-field save with layer
        # targeting): default = provenance auto-routing
        target = None
        if len(writable) > 1:
            chosen = p.string(
                f"target layer [{'/'.join(writable)}] (empty = auto)", "")
            if chosen.strip():
                if chosen.strip() not in writable:
                    io.error(f"no writable layer '{chosen.strip()}'")
                    continue
                target = chosen.strip()
        layer = store.set(path, value, layer=target)
        store.write()
        io.success(f"{path} = {value!r} (layer: {layer})")
        changed += 1
    return changed
