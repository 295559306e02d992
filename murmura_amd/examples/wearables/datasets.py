"""Wearable-sensor dataset loaders: UCI HAR, PAMAP2, PPG-DaLiA.

Fresh implementations against the public dataset formats (the reference
loads the same files: murmura/examples/wearables/datasets.py:12-531).

- UCI HAR: 561 precomputed features per row (X_<split>.txt), 6 activities
  (y_<split>.txt, 1-based), 30 subjects (subject_<split>.txt).
- PAMAP2: Protocol/subjectNNN.dat, 54 space-separated columns per line at
  100 Hz; we keep 40 valid features (heart rate + 13 non-orientation
  channels x 3 IMUs), linearly interpolate NaNs per column, cut 100-sample
  windows with majority activity label over the 12 protocol activities.
- PPG-DaLiA: S<N>/S<N>.pkl with multi-rate wrist signals; we build 1 s
  windows of 32 samples x 6 channels (ACC x3 @32 Hz, BVP @64 Hz decimated,
  EDA + TEMP @4 Hz upsampled), majority activity over 7 activity classes
  (ids 1-7; 0 = transient is dropped).
"""

from __future__ import annotations

import pickle
from pathlib import Path
from typing import List, Optional

import numpy as np
import torch
from torch.utils.data import Dataset


class _ArrayDataset(Dataset):
    """Feature/label/subject triple with tensors materialized up front."""

    def __init__(self, x: np.ndarray, y: np.ndarray, subjects: np.ndarray):
        self.x = torch.as_tensor(x, dtype=torch.float32)
        self.y = torch.as_tensor(y, dtype=torch.long)
        self.subjects = np.asarray(subjects)

    def __len__(self) -> int:
        return self.x.shape[0]

    def __getitem__(self, i: int):
        return self.x[i], self.y[i]


# ------------------------------------------------------------------ UCI HAR
class UCIHARDataset(_ArrayDataset):
    NUM_CLASSES = 6
    NUM_FEATURES = 561

    def __init__(self, data_path: str, split: str = "train",
                 max_samples: Optional[int] = None):
        root = Path(data_path)
        d = root / split
        if not d.is_dir():
            raise FileNotFoundError(f"UCI HAR split dir not found: {d}")
        x = np.loadtxt(d / f"X_{split}.txt")
        y = np.loadtxt(d / f"y_{split}.txt", dtype=int) - 1  # 1-based -> 0-based
        s = np.loadtxt(d / f"subject_{split}.txt", dtype=int)
        if x.ndim == 1:
            x = x[None, :]
            y, s = np.atleast_1d(y), np.atleast_1d(s)
        if max_samples is not None:
            x, y, s = x[:max_samples], y[:max_samples], s[:max_samples]
        super().__init__(x, y, s)


# ------------------------------------------------------------------ PAMAP2
# 12 protocol activities (PAMAP2 activityID -> class index)
PAMAP2_ACTIVITIES = {1: 0, 2: 1, 3: 2, 4: 3, 5: 4, 6: 5, 7: 6, 12: 7, 13: 8,
                     16: 9, 17: 10, 24: 11}
# column layout: 0 timestamp, 1 activityID, 2 heart_rate, then 3 IMUs x 17
# cols (1 temp, 3 acc16g, 3 acc6g, 3 gyro, 3 mag, 4 orientation-invalid)
_PAMAP2_IMU_KEEP = [0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12]  # drop 4 orientation


def _pamap2_feature_cols() -> List[int]:
    cols = [2]  # heart rate
    for imu in range(3):
        base = 3 + imu * 17
        cols.extend(base + k for k in _PAMAP2_IMU_KEEP)
    return cols  # 1 + 39 = 40


def _interp_nan(col: np.ndarray) -> np.ndarray:
    mask = np.isnan(col)
    if mask.all():
        return np.zeros_like(col)
    if mask.any():
        idx = np.arange(len(col))
        col = col.copy()
        col[mask] = np.interp(idx[mask], idx[~mask], col[~mask])
    return col


class PAMAP2Dataset(_ArrayDataset):
    NUM_CLASSES = 12
    WINDOW = 100
    NUM_FEATURES = 40

    def __init__(self, data_path: str, split: str = "train", step: int = 100,
                 max_samples: Optional[int] = None):
        root = Path(data_path)
        proto = root / "Protocol" if (root / "Protocol").is_dir() else root
        files = sorted(proto.glob("subject*.dat"))
        if not files:
            raise FileNotFoundError(f"no PAMAP2 subject*.dat under {proto}")
        cols = _pamap2_feature_cols()
        xs, ys, subs = [], [], []
        for f in files:
            sid = int("".join(ch for ch in f.stem if ch.isdigit()))
            raw = np.loadtxt(f)
            if raw.ndim == 1:
                raw = raw[None, :]
            feats = raw[:, cols]
            for c in range(feats.shape[1]):
                feats[:, c] = _interp_nan(feats[:, c])
            act = raw[:, 1].astype(int)
            for lo in range(0, len(raw) - self.WINDOW + 1, step):
                w_act = act[lo : lo + self.WINDOW]
                vals, counts = np.unique(w_act, return_counts=True)
                major = int(vals[np.argmax(counts)])
                if major not in PAMAP2_ACTIVITIES:
                    continue  # skip transient (0) / non-protocol activities
                xs.append(feats[lo : lo + self.WINDOW])
                ys.append(PAMAP2_ACTIVITIES[major])
                subs.append(sid)
                if max_samples is not None and len(xs) >= max_samples:
                    break
            if max_samples is not None and len(xs) >= max_samples:
                break
        if not xs:
            raise ValueError("PAMAP2: no labeled windows found")
        x = np.stack(xs).reshape(len(xs), -1)  # [N, 100*40]
        super().__init__(x, np.asarray(ys), np.asarray(subs))


# ------------------------------------------------------------------ PPG-DaLiA
class PPGDaLiADataset(_ArrayDataset):
    NUM_CLASSES = 7
    WINDOW = 32  # 1 s at 32 Hz
    NUM_CHANNELS = 6

    def __init__(self, data_path: str, split: str = "train",
                 max_samples: Optional[int] = None):
        root = Path(data_path)
        pkls = sorted(root.glob("**/S*.pkl"))
        if not pkls:
            raise FileNotFoundError(f"no S*.pkl files under {root}")
        xs, ys, subs = [], [], []
        for f in pkls:
            with open(f, "rb") as fh:
                blob = pickle.load(fh, encoding="latin1")
            wrist = blob["signal"]["wrist"]
            acc = np.asarray(wrist["ACC"], dtype=np.float64)  # [T32, 3] @32 Hz
            bvp = np.asarray(wrist["BVP"], dtype=np.float64).reshape(-1)  # @64 Hz
            eda = np.asarray(wrist["EDA"], dtype=np.float64).reshape(-1)  # @4 Hz
            temp = np.asarray(wrist["TEMP"], dtype=np.float64).reshape(-1)  # @4 Hz
            act = np.asarray(blob["activity"], dtype=np.float64).reshape(-1)  # @4 Hz
            sid = int("".join(ch for ch in Path(f).stem if ch.isdigit()) or 0)
            n_sec = min(len(acc) // 32, len(bvp) // 64, len(eda) // 4,
                        len(temp) // 4, len(act) // 4)
            for s in range(n_sec):
                a = act[s * 4 : (s + 1) * 4].astype(int)
                vals, counts = np.unique(a, return_counts=True)
                major = int(vals[np.argmax(counts)])
                if not (1 <= major <= 7):
                    continue  # 0 = transient; ids 1..7 -> classes 0..6
                w = np.empty((self.WINDOW, self.NUM_CHANNELS))
                w[:, 0:3] = acc[s * 32 : (s + 1) * 32]
                w[:, 3] = bvp[s * 64 : (s + 1) * 64 : 2]
                w[:, 4] = np.repeat(eda[s * 4 : (s + 1) * 4], 8)
                w[:, 5] = np.repeat(temp[s * 4 : (s + 1) * 4], 8)
                xs.append(w)
                ys.append(major - 1)
                subs.append(sid)
                if max_samples is not None and len(xs) >= max_samples:
                    break
            if max_samples is not None and len(xs) >= max_samples:
                break
        if not xs:
            raise ValueError("PPG-DaLiA: no labeled windows found")
        x = np.stack(xs).reshape(len(xs), -1)  # [N, 32*6]
        super().__init__(x, np.asarray(ys), np.asarray(subs))


def get_wearable_dataset_info(dataset_type: str) -> dict:
    infos = {
        "uci_har": {"num_classes": 6, "in_features": 561, "num_subjects": 30},
        "pamap2": {"num_classes": 12, "in_features": 4000, "num_subjects": 9},
        "ppg_dalia": {"num_classes": 7, "in_features": 192, "num_subjects": 15},
    }
    key = dataset_type.lower()
    if key not in infos:
        raise ValueError(f"unknown wearable dataset {dataset_type!r}")
    return infos[key]
