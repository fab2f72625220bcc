"""storeui field editor (reference: internal/storeui reduced to a
prompter loop)."""
from clawker_amd.config.schema import Settings
from clawker_amd.iostreams import TestIOStreams
from clawker_amd.storage import Store
from clawker_amd.storeui import edit_store, leaf_fields


def test_leaf_fields_cover_nested_schema():
    fields = leaf_fields(Settings)
    assert "firewall.bypass_max_s" in fields
    assert "gpu.hbm_gb_per_device" in fields
    assert "monitoring.sample_interval_ms" in fields
    assert not any(f.endswith(".") for f in fields)


def test_edit_store_roundtrip(tmp_path, monkeypatch):
    from clawker_amd.storage import Layer
    path = tmp_path / "settings.yaml"
    store = Store(Settings, [Layer("settings", path)])
    fields = leaf_fields(Settings)
    idx = fields.index("firewall.bypass_max_s") + 1
    # select field, enter 900, then finish; force promptability
    io = TestIOStreams(stdin_text=f"{idx}\n900\n\n")
    monkeypatch.setattr(io, "can_prompt", lambda: True)
    changed = edit_store(store, io)
    assert changed == 1
    store2 = Store(Settings, [Layer("settings", path)])
    assert store2.get().firewall.bypass_max_s == 900


def test_edit_store_filter_and_layer_browser(tmp_path, monkeypatch):
    from clawker_amd.storage import Layer
    path = tmp_path / "settings.yaml"
    store = Store(Settings, [Layer("settings", path)])
    # filter to firewall fields, inspect layers of field 1, then finish
    io = TestIOStreams(stdin_text="/bypass\n?1\n\n")
    monkeypatch.setattr(io, "can_prompt", lambda: True)
    changed = edit_store(store, io)
    assert changed == 0
    out = io.out
    assert "bypass_max_s" in out
    assert "by layer" in out


def test_edit_store_layer_targeting(tmp_path, monkeypatch):
    """Two writable layers: the user can target a specific one
    (reference: storeui per-field save with layer targeting)."""
    from clawker_amd.storage import Layer
    g = tmp_path / "global.yaml"
    l = tmp_path / "local.yaml"
    store = Store(Settings, [Layer("global", g, writable=True),
                             Layer("local", l, writable=True)])
    fields = leaf_fields(Settings)
    idx = fields.index("firewall.bypass_max_s") + 1
    # select field, value 600, target layer "global", finish
    io = TestIOStreams(stdin_text=f"{idx}\n600\nglobal\n\n")
    monkeypatch.setattr(io, "can_prompt", lambda: True)
    changed = edit_store(store, io)
    assert changed == 1
    assert "bypass_max_s: 600" in g.read_text()
    assert not l.exists() or "bypass_max_s" not in l.read_text()
