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


def edit_store(store: Store, io: IOStreams, max_rounds: int = 100) -> int:
    """Interactive loop; returns the number of fields changed."""
    p = Prompter(io)
    fields = leaf_fields(store.schema)
    changed = 0
    for _ in range(max_rounds):
        io.print("")
        for i, path in enumerate(fields, 1):
            val = store.get_path(path)
            owner = store.provenance(path) or "defaults"
            io.print(f" {i:2}) {path} = {val!r}  [dim]({owner})[/dim]")
        sel = p.string("field number to edit (empty to finish)", "")
        if not sel.strip():
            break
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
        layer = store.set(path, value)
        store.write()
        io.success(f"{path} = {value!r} (layer: {layer})")
        changed += 1
    return changed
