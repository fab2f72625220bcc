"""Interactive prompts, CI-aware (reference: internal/prompter)."""
from __future__ import annotations

from .errors import ClawkerError
from .iostreams import IOStreams


class Prompter:
    def __init__(self, io: IOStreams):
        self.io = io

    def _require_tty(self, what: str) -> None:
        if not self.io.can_prompt():
            raise ClawkerError(
                f"cannot prompt for {what} (non-interactive); pass the value "
                f"via flags or use --yes")

    def string(self, prompt: str, default: str = "") -> str:
        self._require_tty(prompt)
        suffix = f" [{default}]" if default else ""
        self.io.stderr.write(f"{prompt}{suffix}: ")
        self.io.stderr.flush()
        val = self.io.stdin.readline().strip()
        return val or default

    def confirm(self, prompt: str, default: bool = False) -> bool:
        self._require_tty(prompt)
        suffix = " [Y/n]" if default else " [y/N]"
        self.io.stderr.write(f"{prompt}{suffix}: ")
        self.io.stderr.flush()
        val = self.io.stdin.readline().strip().lower()
        if not val:
            return default
        return val in ("y", "yes")

    def select(self, prompt: str, options: list[str], default: int = 0) -> str:
        self._require_tty(prompt)
        for i, opt in enumerate(options):
            marker = "*" if i == default else " "
            self.io.stderr.write(f" {marker} {i + 1}) {opt}\n")
        self.io.stderr.write(f"{prompt} [1-{len(options)}]: ")
        self.io.stderr.flush()
        val = self.io.stdin.readline().strip()
        if not val:
            return options[default]
        try:
            idx = int(val) - 1
            if 0 <= idx < len(options):
                return options[idx]
        except ValueError:
            if val in options:
                return val
        raise ClawkerError(f"invalid selection: {val}")
