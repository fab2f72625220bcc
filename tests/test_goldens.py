"""Golden-file tests (reference: bundler Dockerfile goldens, firewall
Corefile goldens, storage struct goldens — regenerate with
GOLDEN_UPDATE=1 pytest tests/test_goldens.py)."""
import json
import os
from pathlib import Path

import pytest

GOLDEN_DIR = Path(__file__).resolve().parent / "golden"


def check_golden(name: str, content: str):
    GOLDEN_DIR.mkdir(exist_ok=True)
    path = GOLDEN_DIR / name
    if os.environ.get("GOLDEN_UPDATE") or not path.exists():
        path.write_text(content)
        if not os.environ.get("GOLDEN_UPDATE") and not path.exists():
            pytest.fail(f"golden {name} created; re-run")
        return
    assert content == path.read_text(), (
        f"golden mismatch for {name}; GOLDEN_UPDATE=1 to regenerate")


@pytest.fixture
def cfg(isolated_env, tmp_path, monkeypatch):
    root = tmp_path / "gproj"
    root.mkdir()
    (root / ".clawker.yaml").write_text(
        "project: golden\n"
        "build:\n  stacks: [python]\n  packages: [jq]\n"
        "  steps:\n    - echo custom-step\n"
        "agent:\n  harness: echo\n")
    from clawker_amd.config import load_config
    return load_config(root)


def _stable(script: str) -> str:
    # strip host-specific uid/gid so the golden is machine-independent
    import re
    return re.sub(r"-u \d+ -g \d+", "-u UID -g GID",
                  re.sub(r"-g \d+ (\w+) 2>", r"-g GID \1 2>", script))


def test_base_script_golden(cfg):
    from clawker_amd.bundle import load_harness
    from clawker_amd.bundler import Builder
    from clawker_amd.engine import Engine
    eng = Engine.__new__(Engine)   # script gen needs no engine state
    b = Builder(cfg, eng)
    harness = load_harness("echo", cfg.project_root)
    check_golden("base_script.sh", _stable(b.base_script(harness)))


def test_harness_script_golden(cfg):
    from clawker_amd.bundle import load_harness
    from clawker_amd.bundler import Builder
    from clawker_amd.engine import Engine
    eng = Engine.__new__(Engine)
    b = Builder(cfg, eng)
    harness = load_harness("echo", cfg.project_root)
    check_golden("harness_script.sh", b.harness_script(harness))


def test_policy_snapshot_golden(isolated_env):
    from clawker_amd.config.schema import EgressRule
    from clawker_amd.firewall import EgressRulesStore
    from clawker_amd.firewall.policy import compile_policy
    EgressRulesStore().add([
        EgressRule(dst="api.anthropic.com", proto="tls", port=443),
        EgressRule(dst="claude.ai", proto="tls", port=443,
                   deny_paths=["/public", "/share"]),
        EgressRule(dst="github.com", proto="tcp", port=22),
    ])
    pol = compile_policy()
    pol["generated"] = 0   # stable
    check_golden("policy.json", json.dumps(pol, indent=1, sort_keys=True))


def test_claude_harness_floor_golden(isolated_env):
    """Pin the shipped claude egress floor (reference: the 13-domain floor
    with claude.ai path-deny rules is a security contract)."""
    from clawker_amd.bundle import load_harness
    h = load_harness("claude")
    floor = sorted(f"{r.dst}:{r.proto}:{r.port}:deny={','.join(r.deny_paths)}"
                   for r in h.egress)
    check_golden("claude_egress_floor.txt", "\n".join(floor) + "\n")
    assert any("claude.ai" in f and "/share" in f for f in floor)


def test_config_json_schemas_generated():
    """gen_docs emits JSON schemas for clawker.yaml + settings.yaml
    (reference: cmd/gen-docs JSON-schema generation)."""
    import json
    from pathlib import Path
    repo = Path(__file__).resolve().parent.parent
    for name, top in (("clawker", "agent"), ("settings", "control_plane")):
        sch = json.loads((repo / "docs/schema" / f"{name}.schema.json").read_text())
        assert sch["$schema"].startswith("http://json-schema.org/")
        assert top in sch["properties"]
        assert sch["properties"][top]["type"] == "object"
    # nested defaults survive: gpu.count default present in project schema
    sch = json.loads((repo / "docs/schema/clawker.schema.json").read_text())
    gpu = sch["properties"]["gpu"]["properties"]
    assert "count" in gpu
