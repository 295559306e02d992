"""Reference-YAML ingestion: verbatim configs from the reference repo load
and run (BASELINE.json names the YAML schema an explicit compatibility
surface; SURVEY.md §5.6: extend additively without breaking existing keys).

The YAMLs under tests/data/ref_configs/ are byte-for-byte copies of
/root/reference/experiments/paper/* configs (including dmtt/03_dmtt.yaml with
its ZMQ-era ``distributed:`` block). The real datasets cannot be downloaded
in this environment, so the run-one-round checks substitute a synthetic
shard of the same tensor shape — schema parsing, factory dispatch, parameter
mapping and the round loop are all exercised with the configs unmodified.
"""

import warnings
from pathlib import Path

import pytest
import torch

from murmura_amd.config.loader import load_config
from murmura_amd.core.network import Network
from murmura_amd.utils import factories

REF_DIR = Path(__file__).parent / "data" / "ref_configs"
ALL_CONFIGS = sorted(REF_DIR.glob("*.yaml"))


def test_ref_config_dir_present():
    assert len(ALL_CONFIGS) >= 4


@pytest.mark.parametrize("path", ALL_CONFIGS, ids=lambda p: p.stem)
def test_verbatim_reference_yaml_loads(path):
    """Every copied reference YAML must validate, including ZMQ-era
    distributed keys (transport, startup_grace_s, ...)."""
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")  # legacy-key warnings are expected
        cfg = load_config(path)
    assert cfg.topology.num_nodes > 0
    assert cfg.experiment.rounds > 0


def test_zmq_legacy_keys_warn_and_are_ignored():
    cfg_path = REF_DIR / "03_dmtt.yaml"
    with pytest.warns(UserWarning, match="ZMQ-era"):
        cfg = load_config(cfg_path)
    # round_duration_s maps through to the straggler budget
    assert cfg.distributed.round_duration_s == 120.0
    assert cfg.distributed.transport == "ipc"  # accepted, unused


def _shrink_and_substitute(cfg, num_nodes=4, rounds=1):
    """Synthetic shape-alike in place of the non-downloadable dataset, fewer
    nodes/rounds so one round runs in seconds; everything else verbatim."""
    d = cfg.model_dump()
    d["experiment"]["rounds"] = rounds
    d["experiment"]["verbose"] = False
    d["topology"]["num_nodes"] = num_nodes
    in_features = cfg.model.params.get("input_dim", 561)
    num_classes = cfg.model.params.get("num_classes", 6)
    d["data"] = {
        "adapter": "synthetic",
        "params": {
            "num_samples": 40 * num_nodes,
            "num_features": in_features,
            "num_classes": num_classes,
        },
    }
    d["training"]["batch_size"] = 16
    from murmura_amd.config.schema import Config

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        return Config(**d)


@pytest.mark.parametrize(
    "name", ["uci_har_evidential_trust.yaml", "krum_gaussian_20pct.yaml",
             "pamap2_evidential_trust_gaussian_20pct.yaml",
             "ppg_dalia_balance_directed_deviation_30pct.yaml"]
)
def test_reference_config_runs_one_round(name):
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        cfg = _shrink_and_substitute(load_config(REF_DIR / name))
        mf = factories.build_model_factory(cfg)
        net = Network.from_config(
            cfg,
            mf,
            factories.build_dataset_adapter(cfg),
            factories.build_aggregator_factory(cfg, mf),
            device=torch.device("cpu"),
            criterion_factory=factories.build_criterion_factory(cfg),
            evidential=factories.is_evidential(cfg),
        )
        h = net.train(rounds=1, local_epochs=1, lr=cfg.training.lr)
    assert len(h["round"]) == 1
    assert 0.0 <= h["mean_accuracy"][0] <= 1.0


def test_dmtt_reference_config_runs_one_round():
    """03_dmtt.yaml: mobility + dmtt + topology_liar + ZMQ block, verbatim.
    Simulation backend (the distributed DMTT path has its own gloo tests)."""
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        cfg = _shrink_and_substitute(load_config(REF_DIR / "03_dmtt.yaml"))
        assert cfg.dmtt is not None and cfg.mobility is not None
        mf = factories.build_model_factory(cfg)
        net = Network.from_config(
            cfg,
            mf,
            factories.build_dataset_adapter(cfg),
            factories.build_aggregator_factory(cfg, mf),
            device=torch.device("cpu"),
            criterion_factory=factories.build_criterion_factory(cfg),
            evidential=factories.is_evidential(cfg),
        )
        assert net.mobility is not None
        h = net.train(rounds=1, local_epochs=1, lr=cfg.training.lr)
    assert len(h["round"]) == 1


def test_evidential_trust_reference_param_names_map():
    """vacuity_threshold/accuracy_weight/trust_threshold/self_weight from the
    reference configs must reach the aggregator's tau_u/w_a/tau_base/alpha_self."""
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        cfg = load_config(REF_DIR / "uci_har_evidential_trust.yaml")
        agg = factories.build_aggregator_factory(
            cfg, factories.build_model_factory(cfg)
        )(0)
    assert agg.tau_u == 0.5
    assert agg.w_a == 0.7
    assert agg.tau_base == 0.1
    assert agg.alpha_self == 0.6


def test_unknown_aggregation_params_warn_not_fail():
    """The reference ctors take **kwargs and drop unknown params (krum.py:15
    ignores the configs' 'm'); YAML-level leniency must match."""
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        cfg = load_config(REF_DIR / "krum_gaussian_20pct.yaml")
    assert cfg.aggregation.params.get("m") == 5
    with pytest.warns(UserWarning, match="ignoring unknown keys"):
        agg = factories.build_aggregator_factory(
            cfg, factories.build_model_factory(cfg)
        )(0)
    assert agg.num_compromised == 3  # 'f' alias mapped
