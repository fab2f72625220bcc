"""Image build pipeline tests (ns backend: overlay layer builds)."""
import json
from pathlib import Path

import pytest

from conftest import requires_isolation


@pytest.fixture
def ctx(isolated_env, tmp_path):
    root = tmp_path / "bproj"
    root.mkdir()
    (root / ".clawker.yaml").write_text(
        "project: btest\nagent:\n  harness: echo\n"
        "build:\n  steps:\n    - echo project-step > /etc/project-step\n")
    from clawker_amd.config import load_config
    from clawker_amd.engine import Engine
    cfg = load_config(root)
    eng = Engine()
    yield cfg, eng
    for info in eng.list():
        try:
            eng.remove(info.name, force=True)
        except Exception:
            pass
    eng.close()


@requires_isolation
def test_build_layer_commits_cow_residue(ctx):
    cfg, eng = ctx
    from clawker_amd.engine.build import build_layer
    lid = build_layer(eng, "hostfs", "echo hi > /layer-marker\nmkdir -p /opt/x")
    fs = eng.images.layer_path(lid)
    assert (fs / "layer-marker").read_text().strip() == "hi"
    assert (fs / "opt/x").is_dir()
    # build sandbox runtime residue scrubbed
    assert not (fs / "run/clawker").exists()


@requires_isolation
def test_build_failure_reports_script_tail(ctx):
    cfg, eng = ctx
    from clawker_amd.engine.build import build_layer
    from clawker_amd.errors import EngineError
    with pytest.raises(EngineError, match="boom-marker"):
        build_layer(eng, "hostfs", "echo boom-marker; exit 3")


@requires_isolation
def test_two_stage_project_build_and_run(ctx):
    cfg, eng = ctx
    from clawker_amd.bundler import Builder
    lines = []
    name = Builder(cfg, eng).build(on_progress=lines.append)
    assert name == "clawker-btest:echo"
    assert eng.images.exists("clawker-btest:base")
    assert eng.images.exists("clawker-btest:default")
    meta = eng.images.get(name)
    assert meta.parent == "clawker-btest:base"
    assert meta.cmd[0] == "/bin/sh"          # echo harness CMD

    # rebuild: base is cached (hash unchanged)
    lines2 = []
    Builder(cfg, eng).build(on_progress=lines2.append)
    assert any("up to date" in l for l in lines2)

    # run a sandbox FROM the image: project step + plan scripts visible
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    orch = Orchestrator(cfg, eng)
    sb = "clawker.btest.fromimg"
    orch.run(RunOptions(
        agent="fromimg", name=sb, image=name, autostart=True, firewall=False,
        cmd=["/bin/sh", "-c",
             "cat /etc/project-step; ls /etc/clawker/; cat /etc/clawker/egress-floor.yaml | head -2"]))
    assert eng.wait(sb, timeout_s=30) == 0
    out = eng.logs(sb).decode()
    assert "project-step" in out
    assert "pre-run.sh" in out and "post-init.sh" in out
    assert "harness: echo" in out
    orch.teardown(sb, force=True)


@requires_isolation
def test_boot_plans_execute_from_image(ctx):
    """InitPlan runs once (marker), BootPlan every start (reference:
    init_steps.go / boot_steps.go semantics)."""
    cfg, eng = ctx
    from clawker_amd.bundler import Builder
    from clawker_amd.cmdutil import Factory
    from clawker_amd.controlplane.plans import run_boot_plans
    from clawker_amd.iostreams import TestIOStreams
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    img = Builder(cfg, eng).build()
    orch = Orchestrator(cfg, eng)
    sb = "clawker.btest.plans"
    orch.run(RunOptions(agent="plans", name=sb, image=img, autostart=False,
                        firewall=False,
                        cmd=["/bin/sh", "-c", "cat /tmp/clawker-hooks.log"]))
    f = Factory(io=TestIOStreams())
    with orch.client(sb) as c:
        hello = c.hello()
        assert hello["initialized"] is False
        run_boot_plans(f, sb, c, hello, quiet=True)
        c.agent_ready()
    assert eng.wait(sb, timeout_s=30) == 0
    out = eng.logs(sb).decode()
    # echo harness hooks wrote into /tmp inside the sandbox
    assert "post-init hook executed" in out
    assert "pre-run hook executed" in out
    orch.teardown(sb, force=True)


@requires_isolation
def test_image_rm_refuses_while_in_use(ctx):
    cfg, eng = ctx
    from clawker_amd.bundler import Builder
    from clawker_amd.errors import ConflictError
    from clawker_amd.orchestrator import Orchestrator, RunOptions
    img = Builder(cfg, eng).build()
    orch = Orchestrator(cfg, eng)
    name = "clawker.btest.inuse"
    orch.run(RunOptions(agent="inuse", name=name, image=img, autostart=True,
                        firewall=False, cmd=["sleep", "10"]))
    with pytest.raises(ConflictError):
        eng.remove_image(img)
    assert eng.images.exists(img)
    orch.teardown(name, force=True)
    eng.remove_image(img)          # fine once unused
    assert not eng.images.exists(img)


def test_prune_keeps_fresh_staging_layers(isolated_env):
    from clawker_amd.engine.images import ImageStore
    store = ImageStore()
    lid, fs = store.new_layer_dir()        # fresh tmp- layer (in-flight build)
    (fs / "f").write_text("x")
    removed = store.prune_layers()
    assert (store.root / "layers" / lid).exists()   # survived the prune


@requires_isolation
def test_build_layers_carry_no_build_identity(ctx):
    cfg, eng = ctx
    from clawker_amd.engine.build import build_layer
    lid = build_layer(eng, "hostfs", "echo content > /marker")
    fs = eng.images.layer_path(lid)
    assert (fs / "marker").exists()
    assert not (fs / "etc" / "hostname").exists()
    assert not (fs / "etc" / "hosts").exists()


@requires_isolation
def test_image_save_load_roundtrip(ctx, tmp_path):
    """save -> wipe -> load restores the image byte-for-byte (layer ids
    are content-addressed, so dedup on load is safe)."""
    import tarfile
    cfg, eng = ctx
    from clawker_amd.engine.build import build_image
    build_image(eng, "saveme:latest", "hostfs",
                "echo payload > /saved-marker")
    name = "saveme:latest"
    meta = eng.images.get(name)
    assert meta.layers
    out = eng.images.save(name, tmp_path / "img.tar.gz")
    with tarfile.open(out) as tar:
        assert "manifest.json" in tar.getnames()
    # wipe the image + its layers, then load
    eng.images.remove(name)
    assert not eng.images.exists(name)
    loaded = eng.images.load(out)
    assert loaded.name == name
    assert loaded.layers == meta.layers
    for lid in loaded.layers:
        assert eng.images.layer_path(lid).is_dir()
    # load again (layers present): pure no-op dedup, still registered
    eng.images.load(out, rename="copy:latest")
    assert eng.images.get("copy:latest").layers == meta.layers
    # chained image: save carries the parent chain too
    from clawker_amd.engine.build import build_image as _bi
    _bi(eng, "child:latest", name, "echo two > /two-marker")
    out2 = eng.images.save("child:latest", tmp_path / "chain.tar.gz")
    eng.images.remove("child:latest")
    eng.images.remove(name)
    top = eng.images.load(out2)
    assert top.name == "child:latest" and eng.images.exists(name)
    assert len(eng.images.lowerdirs_for("child:latest")) == 3


def test_image_load_rejects_traversal(isolated_env, tmp_path):
    import io
    import tarfile
    import pytest as _pytest
    from clawker_amd.engine.images import ImageStore
    from clawker_amd.errors import ConflictError
    evil = tmp_path / "evil.tar.gz"
    with tarfile.open(evil, "w:gz") as tar:
        m = json.dumps({"name": "evil:latest", "layers": ["x"]}).encode()
        ti = tarfile.TarInfo("manifest.json"); ti.size = len(m)
        tar.addfile(ti, io.BytesIO(m))
        ti = tarfile.TarInfo("layers/../../etc/cron.d/pwn"); ti.size = 0
        tar.addfile(ti, io.BytesIO(b""))
    with _pytest.raises(ConflictError, match="unsafe"):
        ImageStore().load(evil)


@requires_isolation
def test_container_commit_snapshot_layer(ctx):
    """A sandbox's writes (incl. a deletion whiteout) become a reusable
    image layer; new sandboxes from the committed image see both."""
    cfg, eng = ctx
    from clawker_amd import consts
    from clawker_amd.engine import SandboxSpec
    from clawker_amd.engine.build import build_image, commit_sandbox
    build_image(eng, "cbase:latest", "hostfs",
                "echo original > /victim.txt")
    name = consts.SANDBOX_NAME_PREFIX + "btest.committer"
    eng.create(SandboxSpec(
        name=name, hostname="sbx", autostart=True, netns=True,
        cmd=["/bin/sh", "-c", "echo tooled > /opt/agent-tool; rm /victim.txt"]),
        image="cbase:latest")
    eng.start(name)
    assert eng.wait(name, timeout_s=15) == 0
    meta = commit_sandbox(eng, name, "committed:latest", message="tools")
    assert meta.parent == "cbase:latest" and len(meta.layers) == 1
    # full stack resolves through the parent chain without duplicates
    lowers = eng.images.lowerdirs_for("committed:latest")
    assert len(lowers) == len(set(lowers)) == 3
    assert meta.labels["dev.clawker.commit.message"] == "tools"
    # run from the committed image: the write is there, the delete holds
    name2 = consts.SANDBOX_NAME_PREFIX + "btest.fromcommit"
    eng.create(SandboxSpec(
        name=name2, hostname="sbx", autostart=True, netns=True,
        cmd=["/bin/sh", "-c",
             "cat /opt/agent-tool; test ! -e /victim.txt && echo gone"]),
        image="committed:latest")
    eng.start(name2)
    assert eng.wait(name2, timeout_s=15) == 0
    logs = eng.logs(name2).decode()
    assert "tooled" in logs and "gone" in logs
    eng.remove(name, force=True)
    eng.remove(name2, force=True)


def test_image_load_rejects_escaping_links(isolated_env, tmp_path):
    """Symlink/hardlink members pointing outside the layer tree are
    refused (write-through / read-exposure hardening)."""
    import io
    import tarfile
    import pytest as _pytest
    from clawker_amd.engine.images import ImageStore
    from clawker_amd.errors import ConflictError
    for linkname, typ in (("/etc", tarfile.SYMTYPE),
                          ("../../../../etc/shadow", tarfile.LNKTYPE)):
        evil = tmp_path / "evil-link.tar.gz"
        with tarfile.open(evil, "w:gz") as tar:
            m = json.dumps({"name": "evil:latest", "layers": ["x"]}).encode()
            ti = tarfile.TarInfo("manifest.json"); ti.size = len(m)
            tar.addfile(ti, io.BytesIO(m))
            ti = tarfile.TarInfo("layers/x/fs/link")
            ti.type = typ
            ti.linkname = linkname
            tar.addfile(ti)
        with _pytest.raises(ConflictError, match="unsafe link"):
            ImageStore().load(evil)


@requires_isolation
def test_image_history_shows_chain(ctx):
    cfg, eng = ctx
    from click.testing import CliRunner
    from clawker_amd.cli.root import cli
    from clawker_amd.engine.build import build_image
    build_image(eng, "h1:latest", "hostfs", "echo a > /a")
    build_image(eng, "h2:latest", "h1:latest", "echo b > /b")
    r = CliRunner().invoke(cli, ["image", "history", "h2:latest"])
    if r.exception is not None and not isinstance(r.exception, SystemExit):
        raise r.exception
    assert r.exit_code == 0, r.output
    assert "h2:latest" in r.output and "h1:latest" in r.output
    assert "hostfs" in r.output
