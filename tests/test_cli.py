"""CLI tests through click's CliRunner + subprocess e2e (reference test
strategy: command tests through the Factory seam; e2e through the real
binary — SURVEY.md §4)."""
import json
import os
import subprocess
import sys
import time
from pathlib import Path

import pytest
from click.testing import CliRunner

from conftest import requires_isolation

REPO = Path(__file__).resolve().parent.parent


@pytest.fixture
def proj(isolated_env, tmp_path, monkeypatch):
    root = tmp_path / "cliproj"
    root.mkdir()
    monkeypatch.chdir(root)
    return root


def _invoke(args, cwd=None):
    """Invoke through click; ClawkerErrors map to exit codes like main()."""
    from clawker_amd.cli.root import cli
    from clawker_amd.errors import ClawkerError
    runner = CliRunner()
    r = runner.invoke(cli, args, catch_exceptions=True)
    if r.exception is not None and not isinstance(r.exception, SystemExit):
        if isinstance(r.exception, ClawkerError):
            r.exit_code = r.exception.exit_code
        else:
            raise r.exception
    return r


def test_init_creates_config_and_registers(proj):
    r = _invoke(["init", "--yes", "--name", "My Proj", "--harness", "echo",
                 "--gpus", "2", "--vcs", "github"])
    assert r.exit_code == 0, r.output
    import yaml
    doc = yaml.safe_load((proj / ".clawker.yaml").read_text())
    assert doc["project"] == "my-proj"
    assert doc["agent"]["harness"] == "echo"
    assert doc["gpu"]["count"] == 2
    dsts = [e["dst"] for e in doc["security"]["egress"]]
    assert "github.com" in dsts
    from clawker_amd.project import ProjectRegistry
    assert ProjectRegistry().get("my-proj").root == str(proj)
    # re-init refuses
    r2 = _invoke(["init", "--yes"])
    assert r2.exit_code != 0


def test_version_and_help(isolated_env):
    r = _invoke(["version"])
    assert r.exit_code == 0 and "clawker-amd" in r.output
    r = _invoke(["--help"])
    assert "run" in r.output and "firewall" in r.output


def test_firewall_rules_cli(proj):
    _invoke(["init", "--yes", "--harness", "echo"])
    r = _invoke(["firewall", "add", "api.example.com", "--port", "443",
                 "--deny-path", "/share"])
    assert r.exit_code == 0, r.output
    r = _invoke(["firewall", "list", "--format", "json"])
    rules = json.loads(r.output)
    assert rules[0]["dst"] == "api.example.com"
    assert rules[0]["deny_paths"] == ["/share"]
    # dedupe with path merge
    _invoke(["firewall", "add", "api.example.com", "--deny-path", "/public"])
    rules = json.loads(_invoke(["firewall", "list", "--format", "json"]).output)
    assert len(rules) == 1
    assert set(rules[0]["deny_paths"]) == {"/share", "/public"}
    r = _invoke(["firewall", "remove", "api.example.com"])
    assert r.exit_code == 0
    assert json.loads(_invoke(["firewall", "list", "--format", "json"]).output) == []


def test_settings_get_set(isolated_env):
    r = _invoke(["settings", "get", "firewall.bypass_max_s"])
    assert json.loads(r.output) == 3600
    r = _invoke(["settings", "set", "firewall.bypass_max_s", "1800"])
    assert r.exit_code == 0
    assert json.loads(_invoke(["settings", "get", "firewall.bypass_max_s"]).output) == 1800


def test_alias_crud(isolated_env):
    assert _invoke(["alias", "set", "r8", "run --gpus 8"]).exit_code == 0
    assert "r8 = run --gpus 8" in _invoke(["alias", "list"]).output
    assert _invoke(["alias", "delete", "r8"]).exit_code == 0
    assert _invoke(["alias", "delete", "r8"]).exit_code != 0


@requires_isolation
def test_cli_run_ps_logs_rm_e2e(proj):
    """Full verb flow against real sandboxes."""
    _invoke(["init", "--yes", "--name", "e2e", "--harness", "echo"])
    env = dict(os.environ, PYTHONPATH=str(REPO))

    def clawker(*args, timeout=180):
        return subprocess.run([sys.executable, "-m", "clawker_amd", *args],
                              capture_output=True, text=True, timeout=timeout,
                              cwd=str(proj), env=env)

    # detached run
    r = clawker("run", "-d", "--agent", "w1", "--no-firewall",
                "--", "/bin/sh", "-c", "echo from-w1; sleep 20")
    assert r.returncode == 0, r.stderr
    r = clawker("ps", "--format", "json")
    rows = json.loads(r.stdout)
    assert any(x["name"] == "clawker.e2e.w1" and x["state"] == "running" for x in rows)
    # exec
    r = clawker("exec", "w1", "--", "/bin/echo", "exec-ok")
    assert r.returncode == 0 and "exec-ok" in r.stdout
    # logs
    r = clawker("logs", "w1")
    assert "from-w1" in r.stdout
    # stop + wait + rm
    assert clawker("stop", "w1").returncode == 0
    r = clawker("ps", "-a", "--format", "json")
    assert any(x["name"] == "clawker.e2e.w1" and x["state"] == "exited"
               for x in json.loads(r.stdout))
    assert clawker("rm", "w1").returncode == 0
    r = clawker("ps", "-a", "--format", "json")
    assert not any(x["name"] == "clawker.e2e.w1" for x in json.loads(r.stdout))


@requires_isolation
def test_cli_run_exit_code_propagates(proj):
    _invoke(["init", "--yes", "--name", "ec", "--harness", "echo"])
    env = dict(os.environ, PYTHONPATH=str(REPO))
    r = subprocess.run(
        [sys.executable, "-m", "clawker_amd", "run", "--rm", "--no-firewall",
         "--", "/bin/sh", "-c", "exit 9"],
        capture_output=True, text=True, timeout=180, cwd=str(proj), env=env)
    assert r.returncode == 9, (r.stdout, r.stderr)


@requires_isolation
def test_worktree_cli(proj):
    _invoke(["init", "--yes", "--name", "wt", "--harness", "echo"])
    subprocess.run(["git", "init", "-q", "-b", "main"], cwd=proj, check=True)
    subprocess.run(["git", "-c", "user.email=t@t", "-c", "user.name=t",
                    "commit", "-q", "--allow-empty", "-m", "init"],
                   cwd=proj, check=True)
    r = _invoke(["worktree", "add", "feature/x"])
    assert r.exit_code == 0, r.output
    r = _invoke(["worktree", "list", "--format", "json"])
    rows = json.loads(r.output)
    assert rows[0]["branch"] == "feature/x" and rows[0]["status"] == "ok"
    r = _invoke(["worktree", "remove", "feature/x"])
    assert r.exit_code == 0
    assert json.loads(_invoke(["worktree", "list", "--format", "json"]).output) == []


@requires_isolation
def test_cli_cp_copy_roundtrip(proj):
    """clawker cp: host->sandbox and sandbox->host tar streaming."""
    _invoke(["init", "--yes", "--name", "cptest", "--harness", "echo"])
    env = dict(os.environ, PYTHONPATH=str(REPO))

    def clawker(*args, timeout=180):
        return subprocess.run([sys.executable, "-m", "clawker_amd", *args],
                              capture_output=True, text=True, timeout=timeout,
                              cwd=str(proj), env=env)

    # --user root: writes to / (the echo harness default is non-root)
    r = clawker("run", "-d", "--agent", "c1", "--no-firewall", "-u", "root",
                "--", "/bin/sh", "-c", "echo sandbox-data > /srcfile; sleep 30")
    assert r.returncode == 0, r.stderr
    time.sleep(0.3)
    # sandbox -> host
    r = clawker("cp", "clawker.cptest.c1:/srcfile", str(proj / "out"))
    assert r.returncode == 0, r.stderr
    assert (proj / "out" / "srcfile").read_text().strip() == "sandbox-data"
    # host -> sandbox
    (proj / "payload.txt").write_text("host-data")
    r = clawker("cp", str(proj / "payload.txt"), "c1:/incoming")
    assert r.returncode == 0, r.stderr
    r = clawker("exec", "c1", "--", "/bin/cat", "/incoming/payload.txt")
    assert "host-data" in r.stdout
    clawker("rm", "-f", "c1")


def test_doctor_reports_capabilities(isolated_env):
    r = _invoke(["doctor", "--format", "json"])
    checks = {c["check"]: c for c in json.loads(r.output)}
    assert checks["root"]["ok"]
    assert "native runtime" in checks
    assert checks["openssl"]["ok"]


def test_monitor_exporter_metrics(isolated_env):
    import urllib.request
    from clawker_amd.monitor import exporter
    port = 19193
    exporter.ensure_running(port)
    try:
        body = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/metrics", timeout=5).read().decode()
        assert "clawker_sandboxes_running" in body
        assert "clawker_gpu_busy_percent" in body   # HELP/TYPE always present
        # plugin textfile lane (monitoring-units analog)
        from clawker_amd import consts
        md = consts.state_dir() / "metrics.d"
        md.mkdir(parents=True, exist_ok=True)
        (md / "myunit.prom").write_text("myunit_metric 42\n")
        body = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/metrics", timeout=5).read().decode()
        assert "myunit_metric 42" in body
    finally:
        exporter.stop_running()


def test_init_override_layer_in_subdir(proj):
    """init inside an existing project creates a walk-up override, not a
    new registration (reference: init.go subdirectory behavior)."""
    _invoke(["init", "--yes", "--name", "outer", "--harness", "echo"])
    sub = proj / "nested" / "dir"
    sub.mkdir(parents=True)
    os.chdir(sub)
    try:
        r = _invoke(["init", "--yes", "--harness", "codex"])
        assert r.exit_code == 0, r.output
        assert "override layer" in r.output
        assert (sub / ".clawker.yaml").exists()
        from clawker_amd.config import load_config
        cfg = load_config(sub)
        assert cfg.project_root == proj          # still the outer project
        assert cfg.project_slug == "outer"
        assert cfg.project.agent.harness == "codex"   # override applies
        from clawker_amd.project import ProjectRegistry
        assert len(ProjectRegistry().list_projects()) == 1
    finally:
        os.chdir(proj)


@requires_isolation
def test_cli_create_then_start_attach(proj):
    """create -> start -a: boot plans run and console streams to stdout."""
    _invoke(["init", "--yes", "--name", "csa", "--harness", "echo"])
    env = dict(os.environ, PYTHONPATH=str(REPO))

    def clawker(*args, timeout=180, input=None):
        return subprocess.run([sys.executable, "-m", "clawker_amd", *args],
                              capture_output=True, text=True, timeout=timeout,
                              cwd=str(proj), env=env, input=input)

    r = clawker("create", "--agent", "cs", "--no-firewall", "--no-host-services",
                "--", "/bin/sh", "-c", "echo STARTED-VIA-ATTACH; exit 0")
    assert r.returncode == 0, r.stderr
    assert "clawker.csa.cs" in r.stdout
    r = clawker("ps", "-a", "--format", "json")
    assert any(x["name"] == "clawker.csa.cs" and x["state"] == "created"
               for x in json.loads(r.stdout))
    r = clawker("start", "-a", "cs")
    assert r.returncode == 0, (r.stdout, r.stderr)
    assert "STARTED-VIA-ATTACH" in r.stdout
    clawker("rm", "-f", "cs")


def test_prompt_and_bundle_groups(isolated_env, tmp_path):
    # prompts
    src = tmp_path / "p.md"
    src.write_text("do the thing\n")
    assert _invoke(["prompt", "add", "mytask", str(src)]).exit_code == 0
    assert "mytask" in _invoke(["prompt", "list"]).output
    assert "do the thing" in _invoke(["prompt", "show", "mytask"]).output
    from clawker_amd.cli.prompt import resolve_prompt
    assert resolve_prompt("mytask").read_text() == "do the thing\n"
    assert _invoke(["prompt", "rm", "mytask"]).exit_code == 0
    assert _invoke(["prompt", "show", "mytask"]).exit_code != 0
    # bundle install (user tier overrides embedded floor)
    bdir = tmp_path / "myharness"
    bdir.mkdir()
    (bdir / "harness.yaml").write_text(
        "name: myharness\ncmd: [mytool]\nuser: ''\n")
    (bdir / "evil-link").symlink_to("/etc/passwd")
    r = _invoke(["bundle", "install", str(bdir)])
    assert r.exit_code == 0, r.output
    from clawker_amd import consts
    installed = consts.config_dir() / "harnesses" / "myharness"
    assert (installed / "harness.yaml").is_file()
    assert not (installed / "evil-link").exists()   # symlink sanitized
    from clawker_amd.bundle import load_harness
    h = load_harness("myharness")
    assert h.cmd == ["mytool"]
    out = _invoke(["bundle", "list"]).output
    assert "myharness" in out and "user" in out


@requires_isolation
def test_cli_misc_verbs_smoke(proj):
    """system df/prune, firewall resolve/status, controlplane status,
    image build --no-cache, harness/stack listings."""
    _invoke(["init", "--yes", "--name", "misc", "--harness", "echo"])
    r = _invoke(["system", "df", "--format", "json"])
    kinds = {x["kind"] for x in json.loads(r.output)}
    assert {"images", "sandboxes", "volumes"} <= kinds
    r = _invoke(["harness", "list"])
    assert "claude" in r.output and "echo" in r.output
    r = _invoke(["stack", "list"])
    assert "rocm" in r.output and "python" in r.output
    # build twice: second is cached; --no-cache rebuilds
    r = _invoke(["build", "-q"])
    assert r.exit_code == 0, r.output
    from clawker_amd.engine import Engine
    eng = Engine()
    base1 = eng.images.get("clawker-misc:base").layers
    r = _invoke(["build", "-q"])
    assert eng.images.get("clawker-misc:base").layers == base1   # cached
    r = _invoke(["build", "-q", "--no-cache"])
    assert r.exit_code == 0
    eng.close()
    # firewall resolve + status shapes
    _invoke(["firewall", "add", "resolve.test"])
    import os as _os
    _os.environ["CLAWKER_DNS_STATIC"] = "resolve.test=127.0.0.1"
    r = _invoke(["firewall", "resolve", "resolve.test"])
    out = json.loads(r.output)
    assert out["policy"] == "allowed" and "127.0.0.1" in out["ips"]
    r = _invoke(["firewall", "status"])
    st = json.loads(r.output)
    assert st["backend"] in ("ns", "proc") and st["rules"] >= 1
    # controlplane status without a daemon
    r = _invoke(["controlplane", "status"])
    assert json.loads(r.output)["running"] in (True, False)
    r = _invoke(["system", "prune"])
    assert r.exit_code == 0


@requires_isolation
def test_container_diff(proj):
    _invoke(["init", "--yes", "--name", "dft", "--harness", "echo"])
    env = dict(os.environ, PYTHONPATH=str(REPO))

    def clawker(*args, timeout=180):
        return subprocess.run([sys.executable, "-m", "clawker_amd", *args],
                              capture_output=True, text=True, timeout=timeout,
                              cwd=str(proj), env=env)

    # --user root: mutates / and /etc (the echo harness default is non-root)
    r = clawker("run", "-d", "--agent", "df", "--no-firewall", "-u", "root",
                "--no-host-services", "--", "/bin/sh", "-c",
                "echo x > /newfile; rm /etc/issue 2>/dev/null; sleep 30")
    assert r.returncode == 0, r.stderr
    time.sleep(0.5)
    r = clawker("container", "diff", "df")
    # /newfile exists in no lower layer -> Added (docker diff fidelity)
    assert "A /newfile" in r.stdout, r.stdout
    if "/etc/issue" in r.stdout:
        assert "D /etc/issue" in r.stdout
    clawker("rm", "-f", "df")


@requires_isolation
def test_run_with_volume_flag(proj):
    _invoke(["init", "--yes", "--name", "vfl", "--harness", "echo"])
    env = dict(os.environ, PYTHONPATH=str(REPO))
    host_dir = proj / "shared-data"
    host_dir.mkdir()
    (host_dir / "in.txt").write_text("vol-payload")
    r = subprocess.run(
        [sys.executable, "-m", "clawker_amd", "run", "--rm", "--no-firewall",
         "--no-host-services", "-v", f"{host_dir}:/data",
         "-v", "scratch:/scratch", "--name", "clawker.vfl.custom", "--",
         "/bin/sh", "-c",
         "cat /data/in.txt; echo persisted > /scratch/out.txt"],
        capture_output=True, text=True, timeout=180, cwd=str(proj), env=env)
    assert r.returncode == 0, (r.stdout, r.stderr)
    assert "vol-payload" in r.stdout
    # named volume persisted on the host
    from clawker_amd.engine import Engine
    eng = Engine()
    row = eng.db.get_volume("clawker.user.scratch")
    assert row is not None
    assert (Path(row["path"]) / "out.txt").read_text().strip() == "persisted"
    eng.close()


def test_doctor_collect_sos_bundle(isolated_env, tmp_path, monkeypatch):
    """`doctor --collect` ships a forensic tarball with per-sandbox state."""
    import tarfile
    ws = tmp_path / "sosproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: sostest\n")
    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(ws))
    name = "clawker.sostest.a"
    try:
        orch.run(RunOptions(agent="a", name=name, autostart=True, firewall=False,
                            cmd=["/bin/sh", "-c", "echo sos-console-line"]))
        orch.engine.wait(name, timeout_s=30)
        r = _invoke(["doctor", "--collect", str(tmp_path / "bundles")])
        assert r.exit_code == 0, r.output
        out = Path(r.output.strip().splitlines()[-1])
        assert out.is_file() and out.name.startswith("clawker-sos-")
        with tarfile.open(out) as tar:
            names = tar.getnames()
            assert "doctor.json" in names and "meta.json" in names
            assert "sandboxes.json" in names
            assert f"sandboxes/{name}/status.json" in names
            console = tar.extractfile(f"sandboxes/{name}/console.log").read()
            assert b"sos-console-line" in console
            # no credential material travels
            assert not [n for n in names if "auth" in n or n.endswith(".key")]
    finally:
        orch.teardown(name, force=True)
        orch.close()


def test_changelog_teaser_once_per_version(isolated_env):
    """Version bump -> one teaser from the local CHANGELOG, cursor advances
    (reference: update-notifier + changelog teaser, air-gap redesign)."""
    from clawker_amd import update
    from clawker_amd.iostreams import TestIOStreams
    io = TestIOStreams()
    # first run ever: quiet, cursor written
    assert update.maybe_show_teaser(io) is False
    assert update.maybe_show_teaser(io) is False      # same version: quiet
    # simulate an upgrade: rewind the cursor to an older version
    update._cursor_path().write_text('{"last_version": "0.0.9"}')
    assert update.maybe_show_teaser(io) is True
    err = io.err
    assert "0.0.9 → " in err and "new in this version" in err
    assert update.maybe_show_teaser(io) is False      # shown once
    # teaser content comes from the current version's section
    lines = update.teaser_for("0.1.0")
    assert lines and len(lines) <= 6
    assert update.teaser_for("99.99.99") == []


def test_monitor_extensions_lane(proj, tmp_path):
    """metrics.d textfile-collector verbs (monitoring units analog)."""
    prom = tmp_path / "agent_cost.prom"
    prom.write_text("# HELP agent_cost_usd total spend\nagent_cost_usd 1.25\n")
    r = _invoke(["monitor", "extensions", "--install", str(prom)])
    assert r.exit_code == 0
    r = _invoke(["monitor", "extensions"])
    assert "agent_cost.prom" in r.output and "1" in r.output
    # merged into /metrics by the exporter
    from clawker_amd.monitor.exporter import _metrics_text
    assert "agent_cost_usd 1.25" in _metrics_text()
    r = _invoke(["monitor", "extensions", "--remove", "agent_cost"])
    assert r.exit_code == 0
    r = _invoke(["monitor", "extensions", "--remove", "agent_cost"])
    assert r.exit_code != 0
    # plugin/skill aliases resolve to the bundle group
    r = _invoke(["plugin", "list"])
    assert r.exit_code == 0
    r = _invoke(["skill", "list"])
    assert r.exit_code == 0


def test_exec_stdin_semantics(isolated_env, tmp_path, monkeypatch):
    """Without -i, exec must NOT consume stdin (a parent holding the pipe
    open would block it forever); with -i, piped stdin reaches the cmd."""
    import subprocess
    ws = tmp_path / "exproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: extest\n")
    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(ws))
    name = "clawker.extest.a"
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO) + os.pathsep + env.get("PYTHONPATH", "")
    try:
        orch.run(RunOptions(agent="a", name=name, autostart=True,
                            firewall=False, host_services=False,
                            cmd=["/bin/sleep", "30"]))
        # no -i, stdin = a pipe that NEVER closes: exec must still finish
        r, w = os.pipe()
        p = subprocess.run(
            [sys.executable, "-m", "clawker_amd", "exec", name, "--",
             "/bin/echo", "done-no-stdin"],
            stdin=r, capture_output=True, timeout=20, env=env,
            cwd=ws)
        os.close(r), os.close(w)
        assert p.returncode == 0 and b"done-no-stdin" in p.stdout, p.stderr
        # -i: piped stdin is delivered
        p = subprocess.run(
            [sys.executable, "-m", "clawker_amd", "exec", "-i", name, "--",
             "/bin/cat"],
            input=b"piped-bytes", capture_output=True, timeout=20, env=env,
            cwd=ws)
        assert p.returncode == 0 and b"piped-bytes" in p.stdout, p.stderr
    finally:
        orch.teardown(name, force=True)
        orch.close()


def test_firewall_rotate_ca_and_volume_create(proj):
    _invoke(["init", "--yes", "--name", "rc", "--harness", "echo"])
    from clawker_amd.firewall import mitm
    ca1, _ = mitm.ensure_ca()
    pem1 = ca1.read_bytes()
    r = _invoke(["firewall", "rotate-ca"])
    assert r.exit_code == 0, r.output
    assert mitm.ensure_ca()[0].read_bytes() != pem1     # new CA material
    r = _invoke(["volume", "create", "scratch"])
    assert r.exit_code == 0 and "clawker.user.scratch" in r.output
    r = _invoke(["volume", "ls"])
    assert "scratch" in r.output
    r = _invoke(["firewall", "refresh"])                # reload alias
    assert r.exit_code == 0


def test_system_info(proj):
    r = _invoke(["system", "info"])
    assert r.exit_code == 0, r.output
    d = json.loads(r.output)
    assert d["backend"] in ("ns", "proc")
    assert "sandboxes" in d and "paused" in d["sandboxes"]


def test_exec_detach(isolated_env, tmp_path):
    """exec -d returns immediately with an exec id; the job keeps running
    after the client disconnects and its side effects land."""
    ws = tmp_path / "edproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: edtest\n")
    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(ws))
    name = "clawker.edtest.a"
    try:
        orch.run(RunOptions(agent="a", name=name, autostart=True,
                            firewall=False, host_services=False,
                            cmd=["/bin/sleep", "30"]))
        import time as _t
        r = _invoke(["exec", "-d", name, "--",
                     "/bin/sh", "-c", "sleep 0.3; echo bg > /tmp/detached"])
        assert r.exit_code == 0 and r.output.strip().startswith("d")
        deadline = _t.time() + 10
        found = False
        while _t.time() < deadline and not found:
            code, out, _ = orch.engine.exec(name, ["/bin/cat", "/tmp/detached"])
            found = code == 0 and b"bg" in out
            _t.sleep(0.05)
        assert found
    finally:
        orch.teardown(name, force=True)
        orch.close()


def test_completion_scripts(proj):
    for shell, marker in (("bash", "_clawker_completion"),
                          ("zsh", "#compdef clawker"),
                          ("fish", "complete")):
        r = _invoke(["completion", shell])
        assert r.exit_code == 0 and marker in r.output


def test_volume_inspect(proj):
    _invoke(["volume", "create", "vi"])
    r = _invoke(["volume", "inspect", "vi"])
    assert r.exit_code == 0, r.output
    d = json.loads(r.output)[0]
    assert d["name"] == "clawker.user.vi" and d["exists"] is True
    assert isinstance(d["size_bytes"], int)
    r = _invoke(["volume", "inspect", "nope"])
    assert r.exit_code != 0


def test_wait_timeout_flag(isolated_env, tmp_path):
    ws = tmp_path / "wproj"
    ws.mkdir()
    (ws / ".clawker.yaml").write_text("project: wtest\n")
    from clawker_amd.config import load_config
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(load_config(ws))
    name = "clawker.wtest.a"
    try:
        orch.run(RunOptions(agent="a", name=name, autostart=True,
                            firewall=False, host_services=False,
                            cmd=["/bin/sleep", "30"]))
        r = _invoke(["wait", "--timeout", "0.3", name])
        assert r.exit_code == 124
    finally:
        orch.teardown(name, force=True)
        orch.close()


def test_every_command_help_renders(proj):
    """--help must render for every command and group (wiring sweep:
    a bad decorator or import in any verb fails here, not in the field)."""
    import click
    from clawker_amd.cli.root import cli as root

    def walk(cmd, path):
        r = _invoke(path + ["--help"])
        assert r.exit_code == 0, (path, r.output)
        if isinstance(cmd, click.Group):
            for name, sub in cmd.commands.items():
                walk(sub, path + [name])

    walk(root, [])


def test_home_workspace_refused_noninteractive(proj, monkeypatch):
    """Mounting $HOME (or /) as the workspace is refused without
    interactive confirmation (reference: safety.go home-mount prompt)."""
    _invoke(["init", "--yes", "--name", "hometest", "--harness", "echo"])
    import os as _os
    env = dict(_os.environ, PYTHONPATH=str(REPO), HOME=str(proj.parent))
    r = subprocess.run(
        [sys.executable, "-m", "clawker_amd", "run", "-d", "--agent", "h",
         "--no-firewall", "--workspace-mode", "bind",
         "--", "true"],
        capture_output=True, text=True, timeout=60, cwd=str(proj.parent),
        env=env)
    # cwd IS $HOME here and has no project -> workspace would be None;
    # instead drive via a registered project whose root == HOME
    (proj.parent / ".clawker.yaml").write_text("project: homeproj\n")
    r = subprocess.run(
        [sys.executable, "-m", "clawker_amd", "run", "-d", "--agent", "h2",
         "--no-firewall", "--", "true"],
        capture_output=True, text=True, timeout=60, cwd=str(proj.parent),
        env=env)
    assert r.returncode != 0
    assert "refusing to mount" in (r.stderr + r.stdout)
