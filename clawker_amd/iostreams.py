"""Testable I/O streams (reference: internal/iostreams — TTY detection,
colors honoring NO_COLOR, spinners, testable buffer trio via Test()).
Rendering is delegated to rich, which only this module and the tui package
import (the reference enforces the same boundary: only iostreams imports
lipgloss)."""
from __future__ import annotations

import io
import os
import sys
from contextlib import contextmanager
from typing import IO, Any

from rich.console import Console


class IOStreams:
    def __init__(self, stdin: IO | None = None, stdout: IO | None = None,
                 stderr: IO | None = None, force_tty: bool | None = None):
        self.stdin = stdin or sys.stdin
        self.stdout = stdout or sys.stdout
        self.stderr = stderr or sys.stderr
        self._force_tty = force_tty
        no_color = bool(os.environ.get("NO_COLOR"))
        self.console = Console(
            file=self.stdout, no_color=no_color,
            force_terminal=force_tty if force_tty is not None else None,
            highlight=False, soft_wrap=True)
        self.err_console = Console(
            file=self.stderr, no_color=no_color,
            force_terminal=force_tty if force_tty is not None else None,
            highlight=False, soft_wrap=True)

    # -- capabilities --------------------------------------------------------
    def is_stdin_tty(self) -> bool:
        if self._force_tty is not None:
            return self._force_tty
        try:
            return self.stdin.isatty()
        except (AttributeError, ValueError):
            return False

    def is_stdout_tty(self) -> bool:
        if self._force_tty is not None:
            return self._force_tty
        try:
            return self.stdout.isatty()
        except (AttributeError, ValueError):
            return False

    def is_stderr_tty(self) -> bool:
        if self._force_tty is not None:
            return self._force_tty
        try:
            return self.stderr.isatty()
        except (AttributeError, ValueError):
            return False

    def can_prompt(self) -> bool:
        if os.environ.get("CLAWKER_NO_PROMPT") or os.environ.get("CI"):
            return False
        return self.is_stdin_tty() and self.is_stdout_tty()

    def terminal_size(self) -> tuple[int, int]:
        try:
            sz = os.get_terminal_size(self.stdout.fileno())
            return sz.columns, sz.lines
        except (OSError, ValueError, AttributeError):
            return 80, 24

    # -- output --------------------------------------------------------------
    def print(self, *args: Any, **kw: Any) -> None:
        self.console.print(*args, **kw)

    def eprint(self, *args: Any, **kw: Any) -> None:
        self.err_console.print(*args, **kw)

    def success(self, msg: str) -> None:
        self.eprint(f"[green]✓[/green] {msg}")

    def warn(self, msg: str) -> None:
        self.eprint(f"[yellow]![/yellow] {msg}")

    def error(self, msg: str) -> None:
        self.eprint(f"[red]✗[/red] {msg}")

    @contextmanager
    def spinner(self, text: str):
        if self.is_stdout_tty():
            with self.err_console.status(text):
                yield
        else:
            yield

    # -- pager / markdown / alt screen (reference: iostreams pager +
    # markdown render + alt-screen management) -------------------------------
    def markdown(self, text: str) -> None:
        """Render markdown (help pages, changelog teasers)."""
        from rich.markdown import Markdown
        self.console.print(Markdown(text))

    def page(self, text: str, markdown: bool = False) -> None:
        """Long output through $PAGER on a tty; plain print otherwise
        (PAGER= / NO_PAGER disables, like the reference's pager)."""
        pager = os.environ.get("CLAWKER_PAGER", os.environ.get("PAGER", "less"))
        if (not self.is_stdout_tty() or not pager
                or os.environ.get("NO_PAGER")):
            if markdown:
                self.markdown(text)
            else:
                self.print(text)
            return
        rendered = text
        if markdown:
            from rich.markdown import Markdown
            tmp = Console(file=io.StringIO(), force_terminal=True,
                          width=self.terminal_size()[0])
            tmp.print(Markdown(text))
            rendered = tmp.file.getvalue()
        import subprocess
        env = dict(os.environ)
        env.setdefault("LESS", "-FRX")   # quit-if-one-screen, raw colors
        try:
            subprocess.run([*pager.split()], input=rendered.encode(),
                           env=env, stdout=self.stdout)
        except OSError:
            self.print(rendered)

    @contextmanager
    def alt_screen(self):
        """Full-screen mode with guaranteed restore (the reference
        tracks alt-screen state so raw-mode restore doesn't strand the
        terminal)."""
        if not self.is_stdout_tty():
            yield
            return
        self.stdout.write("\x1b[?1049h\x1b[H")
        self.stdout.flush()
        try:
            yield
        finally:
            self.stdout.write("\x1b[?1049l")
            self.stdout.flush()


class TestIOStreams(IOStreams):
    """Buffer trio for tests (reference: iostreams.Test())."""

    __test__ = False    # not a pytest collection target

    def __init__(self, stdin_text: str = ""):
        self.in_buf = io.StringIO(stdin_text)
        self.out_buf = io.StringIO()
        self.err_buf = io.StringIO()
        super().__init__(self.in_buf, self.out_buf, self.err_buf, force_tty=False)

    @property
    def out(self) -> str:
        return self.out_buf.getvalue()

    @property
    def err(self) -> str:
        return self.err_buf.getvalue()
