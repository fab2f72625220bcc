"""Comment-preserving YAML persistence (VERDICT r01 missing #7;
reference: internal/storage/write.go yaml.Node round trip)."""
import yaml

from clawker_amd.storage.yamledit import update_yaml_text

DOC = """\
# clawker project config — hand edited!
project: myproj   # the slug

agent:
  # which harness drives this agent
  harness: claude
  env:
    FOO: bar

gpu:
  count: 2        # two MI355X per sandbox
"""


def test_scalar_set_preserves_comments():
    data = yaml.safe_load(DOC)
    data["gpu"]["count"] = 4
    out = update_yaml_text(DOC, data)
    assert out is not None
    assert yaml.safe_load(out) == data
    assert "# clawker project config — hand edited!" in out
    assert "# two MI355X per sandbox" in out
    assert "count: 4" in out
    assert "# which harness drives this agent" in out


def test_add_nested_key_keeps_layout():
    data = yaml.safe_load(DOC)
    data["agent"]["workdir"] = "/workspace"
    out = update_yaml_text(DOC, data)
    assert out is not None
    assert yaml.safe_load(out) == data
    assert "# the slug" in out


def test_add_top_level_section():
    data = yaml.safe_load(DOC)
    data["security"] = {"firewall": True}
    out = update_yaml_text(DOC, data)
    assert out is not None
    assert yaml.safe_load(out) == data
    assert "hand edited!" in out


def test_delete_key_removes_block():
    data = yaml.safe_load(DOC)
    del data["agent"]["env"]
    out = update_yaml_text(DOC, data)
    assert out is not None
    assert yaml.safe_load(out) == data
    assert "FOO" not in out
    assert "# which harness drives this agent" in out


def test_unsafe_shapes_fall_back():
    # flow-style docs aren't surgically edited: parse-verify rejects
    doc = "a: {b: 1, c: 2}  # flow\n"
    data = yaml.safe_load(doc)
    data["a"]["b"] = 9
    out = update_yaml_text(doc, data)
    # either correct surgery or safe refusal — never wrong data
    if out is not None:
        assert yaml.safe_load(out) == data


def test_noop_returns_original():
    data = yaml.safe_load(DOC)
    assert update_yaml_text(DOC, data) == DOC


def test_store_set_preserves_comments(tmp_path):
    """End-to-end through Store.set + write."""
    p = tmp_path / "cfg.yaml"
    p.write_text(DOC)
    from clawker_amd.config.config import load_config
    import os
    os.environ.setdefault("CLAWKER_CONFIG_DIR", str(tmp_path / "xdg"))
    from clawker_amd.storage.store import Layer, Store
    from clawker_amd.config.schema import Project
    store = Store(Project, [Layer(name="project", path=p, writable=True)])
    store.set("gpu.count", 8)
    store.write()
    text = p.read_text()
    assert "count: 8" in text
    assert "# two MI355X per sandbox" in text
    assert "hand edited!" in text


def test_list_value_updates_keep_comments():
    """Egress-rule list mutations (the most common `settings set`
    payload) re-render the list block while sibling comments stay."""
    doc = """# top
security:
  firewall: true   # keep on
  egress:
    - {dst: a.test, proto: tls, port: 443}
"""
    data = yaml.safe_load(doc)
    data["security"]["egress"].append(
        {"dst": "b.test", "proto": "ssh", "port": 22})
    out = update_yaml_text(doc, data)
    assert out is not None
    assert yaml.safe_load(out) == data
    assert "# keep on" in out
    assert "# top" in out
