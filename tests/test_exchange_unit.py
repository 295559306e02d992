"""Single-process (world_size=1, gloo) unit tests of the exchange layer."""

import os

import pytest
import torch
import torch.distributed as dist

from murmura_amd.parallel import exchange


@pytest.fixture(scope="module", autouse=True)
def _world1():
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29770")
        dist.init_process_group("gloo", rank=0, world_size=1)
    yield
    # leave the group for other single-process tests


def test_allreduce_mean_world1_is_identity():
    x = torch.randn(100)
    out = exchange.allreduce_mean(x)
    assert torch.allclose(out, x)
    assert out.data_ptr() != x.data_ptr()  # input not aliased


def test_exchange_no_neighbors():
    assert exchange.exchange_with_neighbors(torch.randn(10), []) == {}


def test_symmetrize_wants_world1():
    out = exchange.symmetrize_wants([], 1, torch.device("cpu"))
    assert out == [[]]


def test_chunked_gram_no_neighbors_matches_direct():
    from murmura_amd import ops

    own = torch.randn(1003)
    stacked, g = exchange.exchange_chunked_with_gram(own, [], num_chunks=4)
    assert stacked.shape == (1, 1003)
    expect = ops.gram(own.view(1, -1))
    assert torch.allclose(g, expect, rtol=1e-4)


def test_gram_chunk_accumulation_equals_full():
    """Summing per-chunk Grams over column views equals the full Gram."""
    from murmura_amd import ops

    x = torch.randn(5, 997)
    bounds = [0, 250, 500, 997]
    acc = torch.zeros(5, 5)
    for lo, hi in zip(bounds[:-1], bounds[1:]):
        acc += ops.gram(x[:, lo:hi])
    assert torch.allclose(acc, ops.gram(x), rtol=1e-4, atol=1e-4)
    d2 = ops.sq_dists_from_gram(acc)
    from murmura_amd.ops import reference as ref

    assert torch.allclose(d2, ref.pairwise_sq_dists(x), rtol=1e-3, atol=1e-3)
