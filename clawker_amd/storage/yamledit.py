"""Comment-preserving YAML updates (reference: internal/storage/write.go
round-trips yaml.Node so user comments survive `settings set`; PyYAML
drops comments at the scanner, so we do surgical TEXT edits instead).

Strategy: diff the parsed old document against the new data and apply
minimal line edits — replace scalar values in place (keeping inline
comments), insert new keys under their parent block, delete removed
blocks. Every result is verified by re-parsing: if the surgery does not
reproduce exactly the intended data (flow style, anchors, lists of
mappings, any edge we don't model), the caller falls back to a plain
dump. Comments can therefore never corrupt data — worst case they are
lost, which was the old behavior.
"""
from __future__ import annotations

import re
from typing import Any

import yaml

_KEY_RE = re.compile(r"^(\s*)([^#\s\-][^:]*?):(\s*)(.*?)(\s+#.*)?$")


def _diff(old: Any, new: Any, path: tuple = ()) -> list[tuple] | None:
    """Ops: ('set', path, value) scalar replace; ('add', path, value)
    new key; ('del', path). Returns None when the shapes are not
    surgery-safe (non-dict mutation above leaf level)."""
    if old == new:
        return []
    if not (isinstance(old, dict) and isinstance(new, dict)):
        # leaf (scalar or whole list) replacement
        return [("set", path, new)]
    ops: list[tuple] = []
    for k, v in new.items():
        if k not in old:
            ops.append(("add", path + (k,), v))
        else:
            sub = _diff(old[k], v, path + (k,))
            if sub is None:
                return None
            ops.extend(sub)
    for k in old:
        if k not in new:
            ops.append(("del", path + (k,)))
    return ops


class _Doc:
    """Indentation-based line map of a simple block-style YAML mapping
    document. Only mapping keys are tracked; any structure we cannot
    model makes lookups fail -> fallback."""

    def __init__(self, text: str):
        self.lines = text.split("\n")

    def find_key(self, path: tuple) -> tuple[int, int] | None:
        """Returns (line_idx, indent) of the key line for `path`."""
        stack: list[tuple[int, str]] = []   # (indent, key)
        for idx, line in enumerate(self.lines):
            if not line.strip() or line.lstrip().startswith("#"):
                continue
            m = _KEY_RE.match(line)
            if not m:
                # a list item / flow construct: it can't START a tracked
                # mapping path, but it may belong to a deeper value; skip
                continue
            indent = len(m.group(1))
            key = m.group(2).strip().strip('"\'')
            while stack and stack[-1][0] >= indent:
                stack.pop()
            stack.append((indent, key))
            if tuple(k for _, k in stack) == path:
                return idx, indent
        return None

    def block_end(self, start: int, indent: int) -> int:
        """First line index after the block owned by the key at start."""
        i = start + 1
        while i < len(self.lines):
            ln = self.lines[i]
            if ln.strip() and not ln.lstrip().startswith("#"):
                cur = len(ln) - len(ln.lstrip())
                if cur <= indent:
                    break
            i += 1
        return i

    def child_indent(self, start: int, indent: int) -> int:
        for i in range(start + 1, self.block_end(start, indent)):
            ln = self.lines[i]
            if ln.strip() and not ln.lstrip().startswith("#"):
                return len(ln) - len(ln.lstrip())
        return indent + 2


def _render_scalar(value: Any) -> str | None:
    if isinstance(value, (dict, list)):
        return None
    s = yaml.safe_dump(value, default_flow_style=True).strip()
    if s.endswith("\n..."):
        s = s[:-4].strip()
    return s


def _render_block(key: str, value: Any, indent: int) -> list[str]:
    text = yaml.safe_dump({key: value}, sort_keys=False,
                          default_flow_style=False)
    pad = " " * indent
    return [pad + ln if ln.strip() else ln
            for ln in text.rstrip("\n").split("\n")]


def update_yaml_text(text: str, new_data: dict) -> str | None:
    """Apply `new_data` to the original YAML text preserving comments
    and layout. Returns the new text, or None when surgery is unsafe
    (caller should fall back to a plain dump)."""
    try:
        old = yaml.safe_load(text)
    except yaml.YAMLError:
        return None
    if old is None:
        old = {}
    if not isinstance(old, dict) or not isinstance(new_data, dict):
        return None
    ops = _diff(old, new_data)
    if ops is None:
        return None
    if not ops:
        return text
    doc = _Doc(text)
    for op in ops:
        kind, path = op[0], op[1]
        if kind == "set":
            loc = doc.find_key(path)
            if loc is None:
                return None
            idx, indent = loc
            m = _KEY_RE.match(doc.lines[idx])
            if m is None:
                return None
            rendered = _render_scalar(op[2])
            if rendered is not None and m.group(4):
                # scalar in place; keep any trailing comment
                doc.lines[idx] = (m.group(1) + m.group(2) + ":"
                                  + (m.group(3) or " ") + rendered
                                  + (m.group(5) or ""))
            else:
                # value is (or becomes) a block: replace the whole block
                end = doc.block_end(idx, indent)
                doc.lines[idx:end] = _render_block(path[-1], op[2], indent)
        elif kind == "add":
            parent = path[:-1]
            if parent:
                loc = doc.find_key(parent)
                if loc is None:
                    return None
                pidx, pindent = loc
                ci = doc.child_indent(pidx, pindent)
                end = doc.block_end(pidx, pindent)
                doc.lines[end:end] = _render_block(path[-1], op[2], ci)
            else:
                while doc.lines and not doc.lines[-1].strip():
                    doc.lines.pop()
                doc.lines.extend(_render_block(path[-1], op[2], 0))
        elif kind == "del":
            loc = doc.find_key(path)
            if loc is None:
                return None
            idx, indent = loc
            del doc.lines[idx:doc.block_end(idx, indent)]
    out = "\n".join(doc.lines)
    if not out.endswith("\n"):
        out += "\n"
    # the safety net: surgery must reproduce EXACTLY the intended data
    try:
        if yaml.safe_load(out) != new_data:
            return None
    except yaml.YAMLError:
        return None
    return out
