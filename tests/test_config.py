import json

import pytest

from es_pytorch_amd.config import AttrDict, load_config, merge_override


def test_attrdict_access_and_mutation():
    c = AttrDict({"a": {"b": 1}, "l": [1, {"x": 2}]})
    assert c.a.b == 1
    assert c.l[1].x == 2
    c.a.b = 5
    assert c["a"]["b"] == 5
    c.new = {"k": 1}
    assert c.new.k == 1


def test_load_config(tmp_path):
    p = tmp_path / "c.json"
    p.write_text(json.dumps({"general": {"gens": 3}, "noise": {"std": 0.02}}))
    cfg = load_config(str(p))
    assert cfg.general.gens == 3
    assert cfg.noise.std == 0.02


def test_merge_override():
    base = {"a": {"b": 1, "c": 2}, "d": 3}
    merge_override(base, {"a": {"b": 10}})
    assert base == {"a": {"b": 10, "c": 2}, "d": 3}
    with pytest.raises(KeyError):
        merge_override(base, {"zz": 1})


def test_to_dict_roundtrip():
    d = {"a": {"b": [1, 2]}, "c": "x"}
    assert AttrDict(d).to_dict() == d
