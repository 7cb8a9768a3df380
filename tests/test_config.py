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


def test_every_shipped_config_is_well_formed():
    """Every configs/*.json loads, names a known env, and carries the keys
    the entry scripts read — catches config drift as options evolve."""
    import os

    from es_pytorch_amd.envs import make_batched

    root = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                        "configs")
    files = sorted(f for f in os.listdir(root) if f.endswith(".json"))
    assert len(files) >= 15
    for f in files:
        cfg = load_config(os.path.join(root, f))
        if f == "batch.json":  # sweep driver schema, not a run config
            continue
        if f == "multi_agent.json":  # PursuitTag lives in envs.multiagent
            from es_pytorch_amd.envs.multiagent import BatchedPursuitTag
            assert BatchedPursuitTag(3, "cpu").N_AGENTS >= 2
            continue
        assert cfg.general.policies_per_gen % 2 == 0, f
        assert cfg.noise.tbl_size > 0 and 0 < cfg.noise.std < 1, f
        assert cfg.policy.layer_sizes and cfg.policy.lr > 0, f
        env = make_batched(cfg.env.name, 3, "cpu", max_steps=5)
        assert env.ob_dim > 0, f
