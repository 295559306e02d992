import os
import time

import pytest
import torch

from murmura_amd.utils.checkpoint import load_checkpoint, save_checkpoint
from murmura_amd.utils.device import get_device
from murmura_amd.utils.timing import PhaseTimer


def test_get_device_cpu_fallback():
    d = get_device("auto", 0)
    assert d.type in ("cpu", "cuda")
    assert get_device("cpu").type == "cpu"


def test_phase_timer_disabled_by_default():
    t = PhaseTimer()
    with t.phase("x"):
        pass
    assert t.summary() == {}


def test_phase_timer_enabled(monkeypatch):
    monkeypatch.setenv("MURMURA_TIMING", "1")
    t = PhaseTimer()
    with t.phase("work"):
        time.sleep(0.01)
    with t.phase("work"):
        time.sleep(0.01)
    s = t.summary()
    assert 5.0 < s["work"] < 200.0  # mean ms
    assert t.counts["work"] == 2


def test_checkpoint_atomic_write(tmp_path):
    from torch.utils.data import DataLoader

    from murmura_amd.aggregation import FedAvgAggregator
    from murmura_amd.core.node import Node
    from murmura_amd.data.synthetic import make_synthetic_classification
    from murmura_amd.models import SimpleMLP

    ds = make_synthetic_classification(32, num_features=4, num_classes=2)
    node = Node(0, SimpleMLP(4, 8, 2), DataLoader(ds, batch_size=8),
                DataLoader(ds, batch_size=8), FedAvgAggregator(), torch.device("cpu"))
    p = tmp_path / "a.ckpt"
    save_checkpoint(p, 3, [node], {"round": [0, 1, 2, 3]})
    blob = load_checkpoint(p)
    assert blob["round"] == 3
    assert 0 in blob["nodes"]
    assert not (tmp_path / "a.ckpt.tmp").exists()  # atomic rename cleaned up


def test_device_shard_cpu_roundtrip():
    """DeviceShard works on CPU too (shuffle gather + bounds)."""
    from torch.utils.data import DataLoader, TensorDataset

    from murmura_amd.core.gpu_round import DeviceShard

    x = torch.arange(40, dtype=torch.float32).view(20, 2)
    y = torch.arange(20)
    shard = DeviceShard.from_loader(
        DataLoader(TensorDataset(x, y), batch_size=8), torch.device("cpu"),
        torch.float32,
    )
    assert shard.n == 20
    out_x = torch.empty(16, 2)
    out_y = torch.empty(16, dtype=torch.long)
    g = torch.Generator().manual_seed(0)
    shard.shuffled(out_x, out_y, g)
    # every drawn row is a (x, y)-consistent pair from the shard
    for i in range(16):
        assert out_x[i, 0].item() == 2 * out_y[i].item()
    # same generator state -> same permutation
    g2 = torch.Generator().manual_seed(0)
    out_x2 = torch.empty_like(out_x); out_y2 = torch.empty_like(out_y)
    shard.shuffled(out_x2, out_y2, g2)
    assert torch.equal(out_y, out_y2)
