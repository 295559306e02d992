"""Example dataset loaders, tested against tiny synthetic files written in
the real on-disk formats (LEAF JSON shards, UCI HAR txt, PAMAP2 dat,
PPG-DaLiA pickles)."""

import json
import pickle

import numpy as np
import pytest
import torch

from murmura_amd.examples.leaf.adapter import load_leaf_adapter
from murmura_amd.examples.leaf.datasets import (
    LEAFFEMNISTDataset,
    create_leaf_client_partitions,
)
from murmura_amd.examples.wearables.adapter import load_wearable_adapter
from murmura_amd.examples.wearables.datasets import (
    PAMAP2Dataset,
    PPGDaLiADataset,
    UCIHARDataset,
    get_wearable_dataset_info,
)


# ------------------------------------------------------------------ LEAF
def _write_leaf_femnist(tmp_path, users=4, per_user=6):
    rng = np.random.default_rng(0)
    for split in ["train", "test"]:
        d = tmp_path / split
        d.mkdir(parents=True, exist_ok=True)
        blob = {"users": [], "num_samples": [], "user_data": {}}
        for u in range(users):
            name = f"u{u:03d}"
            blob["users"].append(name)
            blob["num_samples"].append(per_user)
            blob["user_data"][name] = {
                "x": rng.random((per_user, 784)).tolist(),
                "y": rng.integers(0, 62, per_user).tolist(),
            }
        (d / "all_data_0.json").write_text(json.dumps(blob))
    return tmp_path


def test_leaf_femnist_parsing(tmp_path):
    _write_leaf_femnist(tmp_path)
    ds = LEAFFEMNISTDataset(str(tmp_path), split="train")
    assert len(ds) == 24
    x, y = ds[0]
    assert x.shape == (1, 28, 28)
    assert 0 <= y < 62
    assert len(ds.user_indices) == 4


def test_leaf_femnist_max_samples(tmp_path):
    _write_leaf_femnist(tmp_path)
    ds = LEAFFEMNISTDataset(str(tmp_path), split="train", max_samples=10)
    assert len(ds) == 10


def test_leaf_partitions_balanced(tmp_path):
    _write_leaf_femnist(tmp_path, users=6, per_user=5)
    ds = LEAFFEMNISTDataset(str(tmp_path))
    parts = create_leaf_client_partitions(ds, 3, seed=1)
    assert sorted(i for p in parts for i in p) == list(range(30))
    sizes = [len(p) for p in parts]
    assert max(sizes) - min(sizes) <= 5  # user-granular balance


def test_leaf_adapter_end_to_end(tmp_path):
    _write_leaf_femnist(tmp_path)
    adapter = load_leaf_adapter("femnist", str(tmp_path), num_nodes=2)
    assert adapter.get_num_clients() == 2
    sub = adapter.get_client_data(0)
    x, y = sub[0]
    assert x.shape == (1, 28, 28)


# ------------------------------------------------------------------ UCI HAR
def _write_uci_har(tmp_path, n=20):
    rng = np.random.default_rng(1)
    for split in ["train", "test"]:
        d = tmp_path / split
        d.mkdir(parents=True, exist_ok=True)
        np.savetxt(d / f"X_{split}.txt", rng.random((n, 561)))
        np.savetxt(d / f"y_{split}.txt", rng.integers(1, 7, n), fmt="%d")
        np.savetxt(d / f"subject_{split}.txt", rng.integers(1, 5, n), fmt="%d")
    return tmp_path


def test_uci_har_parsing(tmp_path):
    _write_uci_har(tmp_path)
    ds = UCIHARDataset(str(tmp_path), split="train")
    assert len(ds) == 20
    x, y = ds[0]
    assert x.shape == (561,)
    assert 0 <= y < 6  # 1-based labels converted


@pytest.mark.parametrize("method", ["dirichlet", "iid", "natural"])
def test_wearable_adapter_partitions(tmp_path, method):
    _write_uci_har(tmp_path, n=40)
    adapter = load_wearable_adapter(
        "uci_har", str(tmp_path), num_nodes=3, partition_method=method, seed=2
    )
    assert adapter.get_num_clients() >= 3
    allidx = sorted(i for p in adapter.get_client_partitions() for i in p)
    if method != "natural":
        assert allidx == list(range(40))


# ------------------------------------------------------------------ PAMAP2
def test_pamap2_windows(tmp_path):
    rng = np.random.default_rng(2)
    proto = tmp_path / "Protocol"
    proto.mkdir()
    n = 250
    raw = np.zeros((n, 54))
    raw[:, 0] = np.arange(n) * 0.01  # timestamp
    raw[:, 1] = 1  # activity id 1 (class 0)
    raw[100:200, 1] = 0  # transient block -> dropped windows
    raw[:, 2] = 90 + rng.random(n)  # heart rate
    raw[:, 3:] = rng.random((n, 51))
    raw[5:9, 2] = np.nan  # NaN heart-rate run -> interpolated
    np.savetxt(proto / "subject101.dat", raw)
    ds = PAMAP2Dataset(str(tmp_path))
    assert len(ds) >= 1
    x, y = ds[0]
    assert x.shape == (4000,)  # 100 x 40
    assert y.item() == 0
    assert torch.isfinite(x).all()  # NaNs interpolated away


def test_pamap2_activity_mapping():
    from murmura_amd.examples.wearables.datasets import PAMAP2_ACTIVITIES

    assert len(PAMAP2_ACTIVITIES) == 12
    assert sorted(PAMAP2_ACTIVITIES.values()) == list(range(12))


# ------------------------------------------------------------------ PPG-DaLiA
def test_ppg_dalia_windows(tmp_path):
    rng = np.random.default_rng(3)
    secs = 10
    blob = {
        "signal": {
            "wrist": {
                "ACC": rng.random((secs * 32, 3)),
                "BVP": rng.random(secs * 64),
                "EDA": rng.random(secs * 4),
                "TEMP": 30 + rng.random(secs * 4),
            }
        },
        "activity": np.repeat([1, 2, 0, 3, 4, 5, 6, 7, 1, 2], 4).astype(float),
        "subject": "S1",
    }
    d = tmp_path / "S1"
    d.mkdir()
    with open(d / "S1.pkl", "wb") as f:
        pickle.dump(blob, f)
    ds = PPGDaLiADataset(str(tmp_path))
    # 10 seconds, one with activity 0 dropped -> 9 windows
    assert len(ds) == 9
    x, y = ds[0]
    assert x.shape == (192,)  # 32 x 6
    assert 0 <= y < 7


def test_dataset_info():
    assert get_wearable_dataset_info("uci_har")["in_features"] == 561
    assert get_wearable_dataset_info("pamap2")["in_features"] == 4000
    assert get_wearable_dataset_info("ppg_dalia")["in_features"] == 192
    with pytest.raises(ValueError):
        get_wearable_dataset_info("fitbit")


# ---------------------------------------------------------------- configs
def test_example_configs_parse():
    from pathlib import Path

    from murmura_amd.config.loader import load_config

    cfg_dir = Path(__file__).parent.parent / "murmura_amd" / "examples" / "configs"
    files = list(cfg_dir.glob("*.yaml"))
    assert len(files) >= 5
    for f in files:
        cfg = load_config(f)
        assert cfg.experiment.rounds > 0


def test_wearable_config_trains_end_to_end(tmp_path):
    """UCI HAR config through the full factory + simulation pipeline (tiny
    synthetic HAR files)."""
    _write_uci_har(tmp_path, n=60)
    from murmura_amd.cli import _run_simulation
    from murmura_amd.config.schema import Config

    cfg = Config(**{
        "experiment": {"rounds": 2, "verbose": False},
        "topology": {"type": "fully", "num_nodes": 3},
        "aggregation": {"algorithm": "evidential_trust"},
        "training": {"local_epochs": 1, "batch_size": 8, "lr": 0.001},
        "data": {"adapter": "wearables.uci_har",
                 "params": {"data_path": str(tmp_path), "partition_method": "iid"}},
        "model": {"factory": "examples.wearables.har_classifier",
                  "params": {"num_classes": 6}},
    })
    h = _run_simulation(cfg, verbose=False)
    assert len(h["round"]) == 2
    assert h["mean_vacuity"][-1] > 0  # evidential metrics flowing


def test_leaf_celeba_with_images(tmp_path):
    """CelebA loader: JSON shards reference image files; images load + resize."""
    from PIL import Image

    from murmura_amd.examples.leaf.datasets import LEAFCelebADataset

    img_dir = tmp_path / "imgs"
    img_dir.mkdir()
    rng = np.random.default_rng(4)
    names = []
    for i in range(6):
        name = f"img_{i:03d}.jpg"
        arr = (rng.random((50, 40, 3)) * 255).astype(np.uint8)
        Image.fromarray(arr).save(img_dir / name)
        names.append(name)
    d = tmp_path / "train"
    d.mkdir()
    blob = {"users": ["u0", "u1"], "num_samples": [3, 3],
            "user_data": {"u0": {"x": names[:3], "y": [0, 1, 0]},
                          "u1": {"x": names[3:], "y": [1, 1, 0]}}}
    (d / "all_data_0.json").write_text(json.dumps(blob))
    ds = LEAFCelebADataset(str(tmp_path), split="train", images_dir=str(img_dir))
    assert len(ds) == 6
    x, y = ds[0]
    assert x.shape == (3, 84, 84)
    assert 0.0 <= x.min() and x.max() <= 1.0
    assert y in (0, 1)
