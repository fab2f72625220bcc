""".env file parsing for agent.env_file (reference: internal/dotenv)."""
from __future__ import annotations

from pathlib import Path

from .errors import ClawkerError


def parse_env_file(path: Path) -> dict[str, str]:
    """KEY=VALUE lines; '#' comments; single/double quotes stripped;
    `export ` prefix tolerated."""
    if not path.is_file():
        raise ClawkerError(f"env file not found: {path}")
    out: dict[str, str] = {}
    for lineno, raw in enumerate(path.read_text().splitlines(), 1):
        line = raw.strip()
        if not line or line.startswith("#"):
            continue
        if line.startswith("export "):
            line = line[len("export "):]
        if "=" not in line:
            raise ClawkerError(f"{path}:{lineno}: expected KEY=VALUE")
        k, _, v = line.partition("=")
        k = k.strip()
        v = v.strip()
        if len(v) >= 2 and v[0] == v[-1] and v[0] in "\"'":
            v = v[1:-1]
        else:
            # strip trailing comment on unquoted values
            if " #" in v:
                v = v.split(" #", 1)[0].rstrip()
        out[k] = v
    return out
