import json

import pytest
import yaml
from pydantic import ValidationError

from murmura_amd.config import Config, load_config, save_config


def test_defaults():
    c = Config()
    assert c.experiment.seed == 42
    assert c.topology.type == "ring"
    assert c.aggregation.algorithm == "fedavg"
    assert c.backend == "simulation"
    assert c.compute.dtype == "fp32"


def test_reference_style_yaml(tmp_path):
    """A reference-shaped YAML config must parse unchanged (the schema is a
    compatibility surface, reference: config/schema.py)."""
    raw = {
        "experiment": {"name": "exp1", "seed": 42, "rounds": 50},
        "topology": {"type": "fully", "num_nodes": 10, "seed": 12345},
        "aggregation": {"algorithm": "krum", "params": {"f": 2}},
        "attack": {"enabled": True, "type": "gaussian", "percentage": 0.2,
                   "params": {"noise_std": 10.0}},
        "training": {"local_epochs": 2, "batch_size": 32, "lr": 0.01},
        "data": {"adapter": "synthetic", "params": {}},
        "model": {"factory": "models.mlp", "params": {}},
        "backend": "simulation",
    }
    p = tmp_path / "c.yaml"
    p.write_text(yaml.safe_dump(raw))
    c = load_config(p)
    assert c.aggregation.params["f"] == 2
    assert c.attack.percentage == 0.2


def test_extra_forbidden():
    with pytest.raises(ValidationError):
        Config(**{"experiment": {"name": "x", "bogus": 1}})
    with pytest.raises(ValidationError):
        Config(**{"not_a_key": {}})


def test_bad_literals_rejected():
    with pytest.raises(ValidationError):
        Config(**{"topology": {"type": "hypercube"}})
    with pytest.raises(ValidationError):
        Config(**{"aggregation": {"algorithm": "median"}})


def test_dmtt_requires_mobility():
    with pytest.raises(ValidationError):
        Config(**{"dmtt": {}})
    c = Config(**{"dmtt": {}, "mobility": {}})
    assert c.dmtt.budget_B == 5
    assert c.mobility.comm_range == 30.0


def test_attack_percentage_range():
    with pytest.raises(ValidationError):
        Config(**{"attack": {"enabled": True, "percentage": 1.5}})


def test_roundtrip_yaml_and_json(tmp_path):
    c = Config(**{"experiment": {"rounds": 7}, "mobility": {"seed": 3}})
    for name in ["a.yaml", "a.json"]:
        p = tmp_path / name
        save_config(c, p)
        c2 = load_config(p)
        assert c2.experiment.rounds == 7
        assert c2.mobility.seed == 3


def test_unknown_extension(tmp_path):
    p = tmp_path / "c.toml"
    p.write_text("")
    with pytest.raises(ValueError):
        load_config(p)
    with pytest.raises(FileNotFoundError):
        load_config(tmp_path / "missing.yaml")


def test_save_load_roundtrip_with_legacy_keys(tmp_path):
    """A config carrying ZMQ-era keys survives save/load (the compat keys
    are plain fields, so round-tripping preserves them)."""
    import warnings

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        c = Config(**{
            "experiment": {"name": "rt", "rounds": 3},
            "topology": {"type": "ring", "num_nodes": 4},
            "aggregation": {"algorithm": "krum", "params": {"f": 1}},
            "training": {}, "data": {}, "model": {},
            "distributed": {"transport": "tcp", "startup_grace_s": 9.0,
                            "round_duration_s": 30.0},
        })
        p = tmp_path / "rt.yaml"
        save_config(c, p)
        c2 = load_config(p)
    assert c2.distributed.transport == "tcp"
    assert c2.distributed.startup_grace_s == 9.0
    assert c2.distributed.round_duration_s == 30.0
    assert c2.aggregation.params == {"f": 1}
