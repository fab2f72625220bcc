"""Storage engine tests (reference test strategy: SURVEY.md §4 —
in-memory stores via NewFromString, t.TempDir isolation, provenance pins)."""
from dataclasses import dataclass, field
from pathlib import Path

import pytest
import yaml

from clawker_amd.storage import (
    Layer, Store, dataclass_defaults, discover_project_layers, materialize,
    merge_layers,
)
from clawker_amd.storage.store import merge_tags


@dataclass
class Inner:
    x: int = 1
    names: list = field(default_factory=list, metadata={"merge": "union"})


@dataclass
class SchemaT:
    version: int = 1
    title: str = "t"
    inner: Inner = field(default_factory=Inner)
    tags: list = field(default_factory=list)


def test_defaults_layer_materializes_struct_defaults():
    d = dataclass_defaults(SchemaT)
    assert d == {"version": 1, "title": "t", "inner": {"x": 1, "names": []}, "tags": []}


def test_merge_overwrite_and_union_with_provenance():
    tags = merge_tags(SchemaT)
    assert tags == {"inner.names": "union"}
    low = Layer("low", None, data={"title": "a", "inner": {"x": 5, "names": ["n1"]},
                                   "tags": ["t1", "t2"]})
    high = Layer("high", None, data={"inner": {"names": ["n2", "n1"]}, "tags": ["t3"]})
    r = merge_layers([low, high], tags)
    assert r.merged["title"] == "a"
    assert r.merged["inner"]["x"] == 5
    # union: dedupe-preserving append
    assert r.merged["inner"]["names"] == ["n1", "n2"]
    # untagged list: overwrite
    assert r.merged["tags"] == ["t3"]
    assert r.provenance["title"] == "low"
    assert r.provenance["tags"] == "high"
    assert r.provenance["inner.x"] == "low"


def test_store_from_string_and_typed_get():
    s = Store.from_string(SchemaT, "title: hello\ninner:\n  x: 9\n")
    v = s.get()
    assert v.title == "hello"
    assert v.inner.x == 9
    assert v.inner.names == []   # default applied


def test_materialize_ignores_unknown_keys():
    v = materialize(SchemaT, {"title": "x", "bogus": 1, "inner": {"x": 2, "junk": 3}})
    assert v.title == "x" and v.inner.x == 2


def test_write_routing_provenance(tmp_path):
    p1 = tmp_path / "base.yaml"
    p2 = tmp_path / "local.yaml"
    p1.write_text("title: base\ninner:\n  x: 3\n")
    p2.write_text("tags: [a]\n")
    s = Store(SchemaT, [Layer("base", p1), Layer("local", p2)])
    # field owned by base → write routes to base
    assert s.set("inner.x", 7) == "base"
    # new field → highest-priority writable layer
    assert s.set("inner.names", ["z"]) == "local"
    s.write()
    assert yaml.safe_load(p1.read_text())["inner"]["x"] == 7
    assert yaml.safe_load(p2.read_text())["inner"]["names"] == ["z"]
    # defaults-owned fields never route to the virtual layer
    assert s.provenance("version") == "defaults"
    assert s.set("version", 2) == "local"


def test_atomic_write_and_reload(tmp_path):
    p = tmp_path / "cfg.yaml"
    s = Store(SchemaT, [Layer("main", p)])
    s.set("title", "v1")
    s.write()
    s2 = Store(SchemaT, [Layer("main", p)])
    assert s2.get().title == "v1"
    assert not list(tmp_path.glob(".cfg.yaml*"))   # no temp litter


def test_migrations_apply_and_resave(tmp_path):
    p = tmp_path / "cfg.yaml"
    p.write_text("version: 0\ntitle: old\n")

    def mig1(d):
        d["title"] = d.get("title", "") + "-migrated"
        return d

    s = Store(SchemaT, [Layer("main", p)], migrations=[(1, mig1)])
    assert s.get().title == "old-migrated"
    on_disk = yaml.safe_load(p.read_text())
    assert on_disk["version"] == 1 and on_disk["title"] == "old-migrated"
    # second load: migration not re-applied
    s2 = Store(SchemaT, [Layer("main", p)], migrations=[(1, mig1)])
    assert s2.get().title == "old-migrated"


def test_discovery_walkup_dir_wins_over_dotfile(tmp_path):
    root = tmp_path / "proj"
    sub = root / "a" / "b"
    sub.mkdir(parents=True)
    (root / ".clawker").mkdir()
    (root / ".clawker" / "clawker.yaml").write_text("project: p\n")
    (root / ".clawker.yaml").write_text("project: dot\n")   # loses to dir form
    (root / ".clawker" / "clawker.local.yaml").write_text("title: l\n")
    found = discover_project_layers(sub)
    assert [f.name for f in found] == ["clawker.yaml", "clawker.local.yaml"]
    assert found[0].parent.name == ".clawker"


def test_discovery_nearest_wins(tmp_path):
    outer = tmp_path / "outer"
    inner = outer / "inner"
    inner.mkdir(parents=True)
    (outer / ".clawker.yaml").write_text("project: outer\n")
    (inner / ".clawker.yaml").write_text("project: inner\n")
    found = discover_project_layers(inner)
    # nearest last = highest priority
    assert found[-1].parent == inner


def test_invalid_yaml_top_level_rejected(tmp_path):
    from clawker_amd.storage.store import StoreError
    p = tmp_path / "bad.yaml"
    p.write_text("- just\n- a\n- list\n")
    with pytest.raises(StoreError):
        Store(SchemaT, [Layer("main", p)])
    with pytest.raises(StoreError):
        Store.from_string(SchemaT, "[1, 2]")


def test_remove_and_get_path(tmp_path):
    s = Store.from_string(SchemaT, "title: x\ninner:\n  x: 5\n")
    assert s.get_path("inner.x") == 5
    assert s.get_path("inner.missing", "dflt") == "dflt"
    assert s.remove("inner.x") is True
    assert s.remove("inner.x") is False
    assert s.get().inner.x == 1   # default shows through again


def test_write_layer_virtual_errors():
    from clawker_amd.storage.store import StoreError
    s = Store.from_string(SchemaT, "{}")
    with pytest.raises(StoreError):
        s.write_layer("defaults")
    with pytest.raises(StoreError):
        s.set("title", "x", layer="nope")


def test_migration_chain_applies_in_order(tmp_path):
    p = tmp_path / "m.yaml"
    p.write_text("version: 0\ntitle: a\n")
    calls = []

    def m1(d):
        calls.append(1)
        d["title"] += "-1"
        return d

    def m2(d):
        calls.append(2)
        d["title"] += "-2"
        return d

    s = Store(SchemaT, [Layer("main", p)], migrations=[(2, m2), (1, m1)])
    assert calls == [1, 2]          # sorted by version despite declaration order
    assert s.get().title == "a-1-2"
    assert yaml.safe_load(p.read_text())["version"] == 2
