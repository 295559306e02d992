"""LEAF benchmark dataset loaders (FEMNIST / CelebA).

Parses the JSON shard format produced by the LEAF preprocessing pipeline
(reference: murmura/examples/leaf/datasets.py:23-199 reads the same files;
this is a fresh implementation against the public LEAF format):

  data_dir/{train,test}/*.json, each file
    {"users": [...], "num_samples": [...],
     "user_data": {user: {"x": [...], "y": [...]}}}

FEMNIST: x = 784 floats -> 1x28x28 grayscale; 62 classes.
CelebA:  x = image filename (loaded from ``images_dir``, resized to 84x84
         RGB); y = binary attribute.
"""

from __future__ import annotations

import json
import random
from pathlib import Path
from typing import Dict, List, Optional, Tuple

import torch
from torch.utils.data import Dataset


def _read_leaf_split(data_path: Path, split: str) -> Tuple[List[str], Dict[str, dict]]:
    """Collect users and user_data across all JSON shards of a split."""
    split_dir = data_path / split
    if not split_dir.is_dir():
        raise FileNotFoundError(f"LEAF split dir not found: {split_dir}")
    users: List[str] = []
    user_data: Dict[str, dict] = {}
    for f in sorted(split_dir.glob("*.json")):
        blob = json.loads(f.read_text())
        for u in blob["users"]:
            if u not in user_data:
                users.append(u)
            user_data[u] = blob["user_data"][u]
    return users, user_data


class LEAFFEMNISTDataset(Dataset):
    """Flattened LEAF FEMNIST: all users' samples concatenated, with per-user
    index ranges retained for natural partitioning."""

    NUM_CLASSES = 62

    def __init__(
        self,
        data_path: str,
        split: str = "train",
        max_samples: Optional[int] = None,
    ) -> None:
        users, user_data = _read_leaf_split(Path(data_path), split)
        xs, ys = [], []
        self.user_indices: Dict[str, List[int]] = {}
        idx = 0
        for u in users:
            ud = user_data[u]
            take = len(ud["y"])
            if max_samples is not None:
                take = min(take, max(0, max_samples - idx))
            if take == 0:
                break
            xs.extend(ud["x"][:take])
            ys.extend(ud["y"][:take])
            self.user_indices[u] = list(range(idx, idx + take))
            idx += take
        self.x = torch.tensor(xs, dtype=torch.float32).view(-1, 1, 28, 28)
        self.y = torch.tensor(ys, dtype=torch.long)

    def __len__(self) -> int:
        return self.x.shape[0]

    def __getitem__(self, i: int):
        return self.x[i], self.y[i]


class LEAFCelebADataset(Dataset):
    """LEAF CelebA: x entries are image filenames under ``images_dir``;
    images are loaded lazily and resized to 84x84 RGB."""

    NUM_CLASSES = 2
    IMG_SIZE = 84

    def __init__(
        self,
        data_path: str,
        split: str = "train",
        images_dir: Optional[str] = None,
        max_samples: Optional[int] = None,
    ) -> None:
        users, user_data = _read_leaf_split(Path(data_path), split)
        self.images_dir = Path(images_dir) if images_dir else Path(data_path) / "raw" / "img_align_celeba"
        names: List[str] = []
        ys: List[int] = []
        self.user_indices: Dict[str, List[int]] = {}
        idx = 0
        for u in users:
            ud = user_data[u]
            take = len(ud["y"])
            if max_samples is not None:
                take = min(take, max(0, max_samples - idx))
            if take == 0:
                break
            names.extend(ud["x"][:take])
            ys.extend(int(v) for v in ud["y"][:take])
            self.user_indices[u] = list(range(idx, idx + take))
            idx += take
        self.names = names
        self.y = torch.tensor(ys, dtype=torch.long)

    def __len__(self) -> int:
        return len(self.names)

    def __getitem__(self, i: int):
        from PIL import Image

        img = Image.open(self.images_dir / self.names[i]).convert("RGB")
        img = img.resize((self.IMG_SIZE, self.IMG_SIZE))
        x = torch.frombuffer(bytearray(img.tobytes()), dtype=torch.uint8)
        x = x.view(self.IMG_SIZE, self.IMG_SIZE, 3).permute(2, 0, 1).float() / 255.0
        return x, self.y[i]


def create_leaf_client_partitions(
    dataset, num_nodes: int, seed: int = 42
) -> List[List[int]]:
    """Natural user grouping -> nodes: users sorted by sample count, seeded
    shuffle within size ties, then round-robin over nodes so shards are
    size-balanced (reference behavior: examples/leaf/datasets.py:300-377)."""
    users = list(dataset.user_indices.keys())
    rng = random.Random(seed)
    rng.shuffle(users)
    users.sort(key=lambda u: -len(dataset.user_indices[u]))
    parts: List[List[int]] = [[] for _ in range(num_nodes)]
    sizes = [0] * num_nodes
    for u in users:
        tgt = min(range(num_nodes), key=lambda i: sizes[i])
        parts[tgt].extend(dataset.user_indices[u])
        sizes[tgt] += len(dataset.user_indices[u])
    return parts


def load_leaf_dataset(
    dataset_type: str,
    data_path: str,
    split: str = "train",
    max_samples: Optional[int] = None,
    **kwargs,
):
    t = dataset_type.lower()
    if t == "femnist":
        return LEAFFEMNISTDataset(data_path, split=split, max_samples=max_samples)
    if t == "celeba":
        return LEAFCelebADataset(data_path, split=split, max_samples=max_samples, **kwargs)
    raise ValueError(f"unknown LEAF dataset: {dataset_type!r} (femnist|celeba)")
