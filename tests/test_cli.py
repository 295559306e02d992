import json

import yaml
from typer.testing import CliRunner

from murmura_amd.cli import app

runner = CliRunner()


def _write_cfg(tmp_path, **overrides):
    cfg = {
        "experiment": {"name": "cli-test", "rounds": 2, "verbose": False},
        "topology": {"type": "ring", "num_nodes": 3},
        "aggregation": {"algorithm": "fedavg"},
        "training": {"batch_size": 16, "lr": 0.05},
        "data": {"adapter": "synthetic",
                 "params": {"num_samples": 120, "num_features": 10, "num_classes": 3}},
        "model": {"factory": "models.mlp",
                  "params": {"in_features": 10, "hidden": 16, "num_classes": 3}},
        "backend": "simulation",
    }
    cfg.update(overrides)
    p = tmp_path / "cfg.yaml"
    p.write_text(yaml.safe_dump(cfg))
    return p


def test_cli_run_simulation(tmp_path):
    p = _write_cfg(tmp_path)
    out = tmp_path / "hist.json"
    result = runner.invoke(app, ["run", str(p), "--quiet", "--output", str(out)])
    assert result.exit_code == 0, result.output
    hist = json.loads(out.read_text())
    assert len(hist["round"]) == 2


def test_cli_run_with_checkpoint(tmp_path):
    p = _write_cfg(tmp_path)
    ckpt = tmp_path / "c.ckpt"
    r1 = runner.invoke(app, ["run", str(p), "--quiet", "--checkpoint", str(ckpt),
                             "--checkpoint-every", "1"])
    assert r1.exit_code == 0, r1.output
    assert ckpt.exists()
    r2 = runner.invoke(app, ["run", str(p), "--quiet", "--checkpoint", str(ckpt),
                             "--resume"])
    assert r2.exit_code == 0, r2.output


def test_cli_list_components():
    result = runner.invoke(app, ["list-components", "aggregators"])
    assert result.exit_code == 0
    for algo in ["fedavg", "krum", "balance", "sketchguard", "ubar", "evidential_trust"]:
        assert algo in result.output
    bad = runner.invoke(app, ["list-components", "nonsense"])
    assert bad.exit_code == 1


def test_cli_list_all():
    result = runner.invoke(app, ["list-components"])
    assert result.exit_code == 0
    assert "topologies" in result.output and "rccl" in result.output


def test_cli_run_distributed_backend(tmp_path):
    """`murmura run` with backend: distributed spawns N processes via the
    DistributedRunner and returns the same history schema (gloo on CPU)."""
    import json

    import yaml

    cfg = {
        "experiment": {"name": "cli-dist", "seed": 42, "rounds": 2,
                       "verbose": False},
        "topology": {"type": "ring", "num_nodes": 2},
        "aggregation": {"algorithm": "fedavg"},
        "training": {"local_epochs": 1, "batch_size": 16, "lr": 0.05},
        "data": {"adapter": "synthetic",
                 "params": {"num_samples": 80, "num_features": 10,
                            "num_classes": 3}},
        "model": {"factory": "models.mlp",
                  "params": {"in_features": 10, "hidden": 8, "num_classes": 3}},
        "backend": "distributed",
        "distributed": {"comm_backend": "gloo", "master_port": 29661},
    }
    p = tmp_path / "dist.yaml"
    p.write_text(yaml.safe_dump(cfg))
    out = tmp_path / "hist.json"
    result = runner.invoke(app, ["run", str(p), "--quiet", "--output", str(out)])
    assert result.exit_code == 0, result.output
    h = json.loads(out.read_text())
    assert len(h["round"]) == 2


def test_cli_device_override(tmp_path):
    import yaml

    cfg = {
        "experiment": {"name": "dev", "seed": 1, "rounds": 1, "verbose": False},
        "topology": {"type": "ring", "num_nodes": 2},
        "aggregation": {"algorithm": "fedavg"},
        "training": {"batch_size": 8, "lr": 0.05},
        "data": {"adapter": "synthetic",
                 "params": {"num_samples": 48, "num_features": 6,
                            "num_classes": 2}},
        "model": {"factory": "models.mlp",
                  "params": {"in_features": 6, "hidden": 8, "num_classes": 2}},
    }
    p = tmp_path / "c.yaml"
    p.write_text(yaml.safe_dump(cfg))
    result = runner.invoke(app, ["run", str(p), "--quiet", "--device", "cpu"])
    assert result.exit_code == 0, result.output
