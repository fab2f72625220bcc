"""dotenv + host-state staging tests (reference: internal/dotenv,
internal/containerfs)."""
import json
from pathlib import Path

import pytest

from conftest import requires_isolation


def test_parse_env_file(tmp_path):
    from clawker_amd.dotenv import parse_env_file
    p = tmp_path / ".env"
    p.write_text(
        "# comment\n"
        "PLAIN=value\n"
        "export EXPORTED=1\n"
        'QUOTED="hello world"\n'
        "SINGLE='x y'\n"
        "TRAILING=val # note\n"
        "EMPTY=\n")
    env = parse_env_file(p)
    assert env == {"PLAIN": "value", "EXPORTED": "1", "QUOTED": "hello world",
                   "SINGLE": "x y", "TRAILING": "val", "EMPTY": ""}
    from clawker_amd.errors import ClawkerError
    (tmp_path / "bad.env").write_text("NOEQUALS\n")
    with pytest.raises(ClawkerError):
        parse_env_file(tmp_path / "bad.env")
    with pytest.raises(ClawkerError):
        parse_env_file(tmp_path / "missing.env")


def test_stage_host_state_allowlists_and_denies_credentials(tmp_path):
    from clawker_amd.containerfs import stage_host_state
    src_home = tmp_path / "home"
    src_home.mkdir()
    (src_home / "settings.json").write_text(json.dumps(
        {"theme": "dark", "apiKey": "SECRET", "editorMode": "vim"}))
    (src_home / ".gitconfig").write_text(
        "[user]\n\tname = Dev\n\temail = d@x\n[credential]\n\thelper = store\n")
    (src_home / "credentials.json").write_text('{"token": "LEAK"}')
    dest = tmp_path / "vol"
    dest.mkdir()
    written = stage_host_state([
        {"src": str(src_home / "settings.json"), "dst": "settings.json",
         "json_allowlist": ["theme", "editorMode"]},
        {"src": str(src_home / ".gitconfig"), "dst": "gitconfig",
         "filter_keys": ["user.name", "user.email"]},
        {"src": str(src_home / "credentials.json"), "dst": "creds.json"},
    ], dest)
    staged = json.loads((dest / "settings.json").read_text())
    assert staged == {"theme": "dark", "editorMode": "vim"}   # no apiKey
    gitconf = (dest / "gitconfig").read_text()
    assert "name = Dev" in gitconf and "helper" not in gitconf
    # credential files never staged
    assert not (dest / "creds.json").exists()
    assert "creds.json" not in written


@requires_isolation
def test_env_file_and_config_volume_in_run(isolated_env, tmp_path):
    ws = tmp_path / "seproj"
    ws.mkdir()
    (ws / ".env.agent").write_text("FROM_FILE=efv\n")
    (ws / ".clawker.yaml").write_text(
        "project: setest\nagent:\n  harness: echo\n  env_file: .env.agent\n")
    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(ws))
    name = "clawker.setest.a"
    try:
        orch.run(RunOptions(agent="a", name=name, autostart=True, firewall=False,
                            cmd=["/bin/sh", "-c", "echo EF=$FROM_FILE"]))
        assert orch.engine.wait(name, timeout_s=30) == 0
        assert b"EF=efv" in orch.engine.logs(name)
    finally:
        orch.teardown(name, force=True)
        orch.close()


@requires_isolation
def test_harness_config_volume_persists_across_recreate(isolated_env, tmp_path):
    """Config volumes survive sandbox removal (reference: durable agent
    state in named volumes; EnsureConfigVolumes fresh-vs-reused)."""
    ws = tmp_path / "cvproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: cvtest\nagent:\n  harness: claude\n")
    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(ws))
    name = "clawker.cvtest.a"
    # claude harness declares config volume .claude under /home/agent (the
    # hostfs base has no 'agent' user; run as root to write via mount path)
    cmd = ["/bin/sh", "-c", "ls -d /home/agent/.claude && echo tok > /home/agent/.claude/state"]
    try:
        orch.run(RunOptions(agent="a", name=name, autostart=True, firewall=False,
                            cmd=cmd, user="root"))
        assert orch.engine.wait(name, timeout_s=30) == 0, orch.engine.logs(name)
        orch.teardown(name, force=True)
        # recreate: state persists through the named volume
        orch.run(RunOptions(agent="a", name=name, autostart=True, firewall=False,
                            cmd=["/bin/cat", "/home/agent/.claude/state"], user="root"))
        assert orch.engine.wait(name, timeout_s=30) == 0
        assert b"tok" in orch.engine.logs(name)
    finally:
        orch.teardown(name, force=True)
        orch.close()


@requires_isolation
def test_real_claude_harness_runs_in_sandbox(isolated_env, tmp_path):
    """The CI image ships the actual Claude Code CLI: run it (--version)
    inside a firewalled sandbox with the claude harness wiring (config
    volume, managed env, egress floor project)."""
    import shutil as _sh
    if not _sh.which("claude"):
        pytest.skip("claude CLI not in image")
    ws = tmp_path / "clproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: cltest\n")   # harness: claude default
    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(ws))
    name = "clawker.cltest.a"
    try:
        orch.run(RunOptions(agent="a", name=name, autostart=True, firewall=True,
                            user="root", cmd=["claude", "--version"]))
        code = orch.engine.wait(name, timeout_s=60)
        logs = orch.engine.logs(name).decode()
        assert code == 0, logs
        assert "Claude Code" in logs or any(c.isdigit() for c in logs), logs
        # harness config volume mounted
        info = orch.engine.inspect(name)
        import json as _json
        spec = _json.loads((info.rundir / "spec.json").read_text())
        assert any(m["dst"].endswith("/.claude") for m in spec["mounts"])
    finally:
        orch.teardown(name, force=True)
        from clawker_amd.controlplane.client import CPClient
        CPClient(auto_start=False).stop()
        orch.close()


def test_staging_denies_credentials_at_depth(tmp_path):
    """The credential deny-list applies inside staged DIRECTORIES too
    (a nested id_rsa must never reach the config volume)."""
    from clawker_amd.containerfs import stage_host_state
    src = tmp_path / "conf"
    (src / "sub").mkdir(parents=True)
    (src / "ok.txt").write_text("fine")
    (src / "sub" / "id_rsa").write_text("PRIVATE")
    (src / "sub" / "note.txt").write_text("also fine")
    dest = tmp_path / "vol"
    dest.mkdir()
    written = stage_host_state(
        [{"src": str(src), "dst": "conf"}], dest)
    assert written
    assert (dest / "conf" / "conf" / "ok.txt").exists() or \
           (dest / "conf" / "ok.txt").exists()
    found = list(dest.rglob("id_rsa"))
    assert not found, f"credential staged: {found}"
    assert list(dest.rglob("note.txt"))
