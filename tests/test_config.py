from pathlib import Path

from clawker_amd.config import Project, Settings, load_config
from clawker_amd.config.config import slugify
from clawker_amd.storage import Store


def test_project_defaults():
    p = Store.from_string(Project, "").get()
    assert p.agent.harness == "claude"
    assert p.workspace.mode == "bind"
    assert p.security.firewall is True
    assert p.gpu.count == 0 and p.gpu.exclusive is True


def test_settings_defaults():
    s = Store.from_string(Settings, "").get()
    assert s.firewall.enable is True
    assert s.gpu.hbm_gb_per_device == 288
    assert s.monitoring.sample_interval_ms == 1000


def test_egress_rules_union_across_layers():
    from clawker_amd.storage import Layer, merge_layers
    from clawker_amd.storage.store import merge_tags
    low = Layer("harness-floor", None, data={
        "security": {"egress": [{"dst": "api.anthropic.com", "proto": "tls", "port": 443}]}})
    high = Layer("project", None, data={
        "security": {"egress": [{"dst": "github.com", "proto": "tls", "port": 443}]}})
    r = merge_layers([low, high], merge_tags(Project))
    dsts = [e["dst"] for e in r.merged["security"]["egress"]]
    assert dsts == ["api.anthropic.com", "github.com"]


def test_load_config_discovers_project(tmp_path, monkeypatch, isolated_env):
    root = tmp_path / "myproj"
    root.mkdir()
    (root / ".clawker.yaml").write_text("project: my-proj\nagent:\n  harness: codex\n")
    cfg = load_config(root)
    assert cfg.project_root == root
    assert cfg.project_slug == "my-proj"
    assert cfg.project.agent.harness == "codex"
    assert cfg.sandbox_name("ralph") == "clawker.my-proj.ralph"
    assert cfg.image_name() == "clawker-my-proj:codex"


def test_slug_default_from_dirname(tmp_path, isolated_env):
    root = tmp_path / "My Repo"
    root.mkdir()
    (root / ".clawker.yaml").write_text("{}\n")
    cfg = load_config(root)
    assert cfg.project_slug == "my-repo"


def test_slugify():
    assert slugify("Hello World!") == "hello-world"
    assert slugify("") == "project"
