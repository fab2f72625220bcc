"""Version-change changelog teaser (reference: internal/update +
internal/changelog + internal/state — update-notifier with a
Keep-a-Changelog teaser and a persisted cursor, cmd.go:364
printUpdateNotification). This node is air-gapped, so there is no
release endpoint to poll; what survives is the *teaser*: the first CLI
invocation after the installed version changes prints the new version's
changelog highlights once, then advances a cursor in the state dir."""
from __future__ import annotations

import json
import re
from pathlib import Path

from . import __version__, consts

_MAX_TEASER_LINES = 6


def _cursor_path() -> Path:
    return consts.state_dir() / "cli-state.json"


def changelog_path() -> Path | None:
    """CHANGELOG.md shipped next to the package (deployment layout)."""
    p = Path(__file__).resolve().parent.parent / "CHANGELOG.md"
    return p if p.is_file() else None


def teaser_for(version: str) -> list[str]:
    """First ≤6 content lines of the `## <version>` changelog section."""
    p = changelog_path()
    if not p:
        return []
    lines: list[str] = []
    in_section = False
    for line in p.read_text().splitlines():
        if line.startswith("## "):
            if in_section:
                break
            in_section = bool(re.match(rf"##\s+{re.escape(version)}\b", line))
            continue
        if in_section and line.strip():
            lines.append(line.rstrip())
            if len(lines) >= _MAX_TEASER_LINES:
                break
    return lines


def maybe_show_teaser(io) -> bool:
    """Print the teaser once per version change; returns True if shown."""
    path = _cursor_path()
    try:
        state = json.loads(path.read_text())
    except (OSError, ValueError):
        state = {}
    last = state.get("last_version")
    if last == __version__:
        return False
    shown = False
    if last is not None:          # first run ever stays quiet
        teaser = teaser_for(__version__)
        if teaser:
            io.eprint(f"[dim]clawker-amd {last} → {__version__} — new in "
                      f"this version:[/dim]")
            for line in teaser:
                io.eprint(f"[dim]{line}[/dim]")
            shown = True
    state["last_version"] = __version__
    try:
        path.parent.mkdir(parents=True, exist_ok=True)
        path.write_text(json.dumps(state))
    except OSError:
        pass
    return shown
