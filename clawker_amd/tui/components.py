"""Reusable terminal components (reference: internal/tui — BubbleTea
progress trees, panels, tables, kv views shared by every command group).
Python redesign over rich: small composable renderables the CLI verbs
share instead of hand-rolling tables/panels per command."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Iterable

PENDING, RUNNING, DONE, FAILED, SKIPPED = range(5)
_MARKS = {PENDING: "[dim]·[/dim]", RUNNING: "[cyan]»[/cyan]",
          DONE: "[green]✓[/green]", FAILED: "[red]✗[/red]",
          SKIPPED: "[dim]-[/dim]"}


@dataclass
class Step:
    name: str
    state: int = PENDING
    detail: str = ""
    children: list["Step"] = field(default_factory=list)


class ProgressSteps:
    """A tree of named steps with live state marks (the RunProgress
    analog). Render returns a rich renderable; drive with start/done/fail
    by name. Safe to render repeatedly inside rich.live.Live."""

    def __init__(self, names: Iterable[str] = ()):
        self.steps: list[Step] = [Step(n) for n in names]

    def add(self, name: str, parent: str | None = None) -> Step:
        s = Step(name)
        if parent:
            self._find(parent).children.append(s)
        else:
            self.steps.append(s)
        return s

    def _find(self, name: str) -> Step:
        def walk(steps):
            for s in steps:
                if s.name == name:
                    return s
                got = walk(s.children)
                if got:
                    return got
            return None
        s = walk(self.steps)
        if s is None:
            s = self.add(name)
        return s

    def start(self, name: str, detail: str = "") -> None:
        s = self._find(name)
        s.state, s.detail = RUNNING, detail

    def done(self, name: str, detail: str = "") -> None:
        s = self._find(name)
        s.state, s.detail = DONE, detail or s.detail

    def fail(self, name: str, detail: str = "") -> None:
        s = self._find(name)
        s.state, s.detail = FAILED, detail or s.detail

    def skip(self, name: str) -> None:
        self._find(name).state = SKIPPED

    @property
    def failed(self) -> bool:
        def walk(steps):
            return any(s.state == FAILED or walk(s.children) for s in steps)
        return walk(self.steps)

    def __rich__(self):
        from rich.tree import Tree
        root = Tree("", hide_root=True)
        def emit(node, steps):
            for s in steps:
                label = f"{_MARKS[s.state]} {s.name}"
                if s.detail:
                    label += f" [dim]{s.detail}[/dim]"
                emit(node.add(label), s.children)
        emit(root, self.steps)
        return root


def kv_panel(title: str, data: dict, *, width: int | None = None):
    """Aligned key/value panel (the tui panel/field-browser analog)."""
    from rich.panel import Panel
    from rich.table import Table
    t = Table(box=None, show_header=False, pad_edge=False, padding=(0, 1))
    t.add_column(style="bold", no_wrap=True)
    t.add_column(overflow="fold")
    for k, v in data.items():
        t.add_row(str(k), _fmt(v))
    return Panel(t, title=title, title_align="left", width=width)


def plain_table(columns: Iterable[str], rows: Iterable[Iterable[Any]]):
    """Headered borderless table — the shared listing look of ps/ls
    verbs (tui table component analog)."""
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for c in columns:
        t.add_column(str(c))
    for r in rows:
        t.add_row(*[_fmt(v) for v in r])
    return t


def _fmt(v: Any) -> str:
    if v is None or v == "":
        return "-"
    if isinstance(v, bool):
        return "yes" if v else "no"
    if isinstance(v, (list, tuple)):
        return ", ".join(str(x) for x in v) or "-"
    if isinstance(v, dict):
        return ", ".join(f"{k}={x}" for k, x in v.items()) or "-"
    return str(v)
