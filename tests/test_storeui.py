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
