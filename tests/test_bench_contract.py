"""bench.py driver-contract tests (CPU): presets build valid configs and the
JSON line carries the required fields."""

import json
import os
import subprocess
import sys

import pytest


def _args(preset):
    import importlib.util

    spec = importlib.util.spec_from_file_location(
        "benchmod", os.path.join(os.path.dirname(__file__), "..", "bench.py")
    )
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


@pytest.mark.parametrize("preset", [1, 2, 3, 4, 5])
def test_presets_build_valid_configs(preset):
    mod = _args(0)

    class A:
        pass

    a = A()
    for k, v in dict(preset=preset, algo="fedavg", topology=None, attack="none",
                     model="resnet18", sketch_wire=False, mobility=False,
                     dmtt=False, dtype="bf16", shard=256, batch_size=16,
                     local_epochs=1, no_eval=False, gpus=2, steps=2,
                     warmup=1).items():
        setattr(a, k, v)
    a = mod.apply_preset(a)
    cfg = mod.build_config(a, 2)
    assert cfg.topology.num_nodes == 2
    if preset == 5:
        assert cfg.dmtt is not None and cfg.mobility is not None
    if preset == 4:
        assert cfg.distributed.sketch_wire_mode


def test_bench_json_contract_cpu():
    """Default invocation emits one JSON line with the driver's fields."""
    env = dict(os.environ)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--model", "mlp", "--dtype", "fp32", "--shard", "256"],
        capture_output=True, text=True, timeout=300,
        cwd=os.path.join(os.path.dirname(__file__), ".."), env=env,
    )
    assert out.returncode == 0, out.stderr[-500:]
    line = [l for l in out.stdout.splitlines() if l.startswith('{"metric"')][0]
    d = json.loads(line)
    for field in ["metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                  "dtype", "data", "config"]:
        assert field in d
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["data"] == "synthetic"
