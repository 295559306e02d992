import numpy as np

from murmura_amd.data.partitioners import (
    dirichlet_partition,
    iid_partition,
    natural_partition,
)


def test_iid_covers_all_indices():
    parts = iid_partition(100, 7, seed=1)
    allidx = sorted(i for p in parts for i in p)
    assert allidx == list(range(100))
    sizes = [len(p) for p in parts]
    assert max(sizes) - min(sizes) <= 1


def test_iid_deterministic():
    assert iid_partition(50, 3, seed=9) == iid_partition(50, 3, seed=9)


def test_dirichlet_covers_all_and_min_samples():
    labels = np.random.default_rng(0).integers(0, 5, size=500).tolist()
    parts = dirichlet_partition(labels, 8, alpha=0.1, seed=4)
    allidx = sorted(i for p in parts for i in p)
    assert allidx == list(range(500))
    assert all(len(p) >= 2 for p in parts)


def test_dirichlet_skew_increases_with_small_alpha():
    labels = (list(range(4)) * 250)
    skewed = dirichlet_partition(labels, 4, alpha=0.05, seed=2)
    uniform = dirichlet_partition(labels, 4, alpha=100.0, seed=2)

    def class_imbalance(parts):
        imb = 0.0
        for p in parts:
            counts = np.bincount([labels[i] for i in p], minlength=4)
            frac = counts / max(1, counts.sum())
            imb += float(frac.max())
        return imb / len(parts)

    assert class_imbalance(skewed) > class_imbalance(uniform)


def test_natural_partition_groups_by_subject():
    sids = [0, 0, 1, 2, 1, 2, 2]
    parts = natural_partition(sids)
    assert parts == [[0, 1], [2, 4], [3, 5, 6]]
    capped = natural_partition(sids, max_clients=2)
    assert len(capped) == 2
