"""hipGraph round execution: captured epoch/eval must match the eager path."""

import os

import pytest
import torch
from torch.utils.data import DataLoader

pytestmark = pytest.mark.gpu


def _make_node(graphs: bool):
    from murmura_amd.aggregation import FedAvgAggregator
    from murmura_amd.core.node import Node
    from murmura_amd.data.synthetic import make_synthetic_classification
    from murmura_amd.models import SimpleMLP

    torch.manual_seed(7)
    ds = make_synthetic_classification(512, num_features=16, num_classes=4, seed=3)
    node = Node(
        0,
        SimpleMLP(16, 32, 4),
        DataLoader(ds, batch_size=64, shuffle=True,
                   generator=torch.Generator().manual_seed(1), drop_last=True),
        DataLoader(ds, batch_size=64),
        FedAvgAggregator(),
        torch.device("cuda:0"),
        model_factory=lambda: SimpleMLP(16, 32, 4),
    )
    return node


def test_graph_eval_matches_eager_eval():
    node = _make_node(graphs=True)
    res_graph = node.evaluate()
    os.environ["MURMURA_NO_GRAPHS"] = "1"
    try:
        res_eager = node.evaluate()
    finally:
        del os.environ["MURMURA_NO_GRAPHS"]
    assert abs(res_graph["accuracy"].item() - res_eager["accuracy"].item()) < 1e-3
    assert abs(res_graph["loss"].item() - res_eager["loss"].item()) < 1e-3


def test_graph_training_learns():
    node = _make_node(graphs=True)
    acc0 = node.evaluate()["accuracy"].item()
    for r in range(5):
        stats = node.local_train(epochs=1, lr=0.1, round_num=r)
        assert stats["num_batches"] == 8  # 512 / 64
    acc1 = node.evaluate()["accuracy"].item()
    assert acc1 > max(acc0, 0.8)


def test_graph_replay_sees_aggregated_state():
    """Graph-captured addresses must stay valid after aggregation overwrites
    the flat buffer in place."""
    node = _make_node(graphs=True)
    node.local_train(epochs=1, lr=0.1)
    acc_trained = node.evaluate()["accuracy"].item()
    # zero the whole state through the aggregation apply path
    node.set_state(torch.zeros_like(node.store.flat))
    res = node.evaluate()  # replays the SAME eval graph
    assert abs(res["accuracy"].item() - acc_trained) > 1e-6 or acc_trained < 0.5
    # a zeroed MLP predicts uniformly -> loss == ln(num_classes)
    import math

    assert abs(res["loss"].item() - math.log(4)) < 1e-2


def test_graph_and_eager_training_similar_quality():
    node_g = _make_node(graphs=True)
    for r in range(3):
        node_g.local_train(epochs=1, lr=0.1, round_num=r)
    acc_g = node_g.evaluate()["accuracy"].item()

    os.environ["MURMURA_NO_GRAPHS"] = "1"
    try:
        node_e = _make_node(graphs=False)
        for r in range(3):
            node_e.local_train(epochs=1, lr=0.1, round_num=r)
        acc_e = node_e.evaluate()["accuracy"].item()
    finally:
        del os.environ["MURMURA_NO_GRAPHS"]
    # different shuffle orders -> not bitwise; both must learn
    assert acc_g > 0.8 and acc_e > 0.8


def test_evidential_eval_graph():
    from murmura_amd.aggregation import EvidentialTrustAggregator
    from murmura_amd.core.node import Node
    from murmura_amd.data.synthetic import make_synthetic_classification
    from murmura_amd.models import EvidentialHARClassifier

    torch.manual_seed(3)
    ds = make_synthetic_classification(256, num_features=561, num_classes=6, seed=5)
    node = Node(
        0,
        EvidentialHARClassifier(),
        DataLoader(ds, batch_size=64, drop_last=True),
        DataLoader(ds, batch_size=64),
        EvidentialTrustAggregator(),
        torch.device("cuda:0"),
        evidential=True,
    )
    res = node.evaluate()
    for k in ["vacuity", "entropy", "strength", "accuracy", "loss"]:
        assert torch.isfinite(res[k]), k
    assert 0.0 < res["vacuity"].item() <= 1.0


def test_evidential_train_graph_learns_and_anneals():
    """Evidential models train through the captured graph: the KL weight is a
    device tensor the replay reads, so annealing still works across rounds."""
    from murmura_amd.aggregation import EvidentialTrustAggregator
    from murmura_amd.core.node import Node
    from murmura_amd.data.synthetic import make_synthetic_classification
    from murmura_amd.models import EvidentialHARClassifier, get_evidential_loss

    torch.manual_seed(11)
    ds = make_synthetic_classification(512, num_features=561, num_classes=6, seed=7)
    node = Node(
        0,
        EvidentialHARClassifier(),
        DataLoader(ds, batch_size=64, shuffle=True,
                   generator=torch.Generator().manual_seed(2), drop_last=True),
        DataLoader(ds, batch_size=64),
        EvidentialTrustAggregator(),
        torch.device("cuda:0"),
        criterion=get_evidential_loss(6, total_rounds=20),
        evidential=True,
    )
    acc0 = node.evaluate()["accuracy"].item()
    losses = []
    for r in range(6):
        stats = node.local_train(epochs=1, lr=0.01, round_num=r)
        losses.append(stats["loss"])
    assert node._train_graph is not None  # graph path engaged
    acc1 = node.evaluate()["accuracy"].item()
    assert acc1 > max(acc0, 0.5)
    # kl weight actually annealed on-device
    assert node._train_graph.kl_weight.item() > 0.0


def test_async_eval_matches_sync_eval():
    """Side-stream async evaluation must produce the same metrics as the
    synchronous path on the same state."""
    from murmura_amd.core.async_eval import AsyncEvaluator

    node = _make_node(graphs=True)
    node.local_train(epochs=1, lr=0.1)
    sync_res = node.evaluate()
    ev = AsyncEvaluator(node)
    h = ev.launch(0)
    res = h.resolve()
    assert abs(res["accuracy"].item() - sync_res["accuracy"].item()) < 1e-3
    assert abs(res["loss"].item() - sync_res["loss"].item()) < 1e-3
    # mutate the live state AFTER launching: the handle must still report the
    # snapshot's metrics (launch before mutation)
    h2 = ev.launch(1)
    node.set_state(torch.zeros_like(node.store.flat))
    res2 = h2.resolve()
    assert abs(res2["accuracy"].item() - sync_res["accuracy"].item()) < 1e-3


def test_async_eval_pipeline_in_round_loop():
    import torch.distributed as dist

    from murmura_amd.config.schema import Config
    from murmura_amd.parallel.node_process import FLRoundLoop

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29750")
        dist.init_process_group(backend="nccl", rank=0, world_size=1)
    cfg = Config(**{
        "experiment": {"rounds": 4, "verbose": False},
        "topology": {"type": "fully", "num_nodes": 1},
        "aggregation": {"algorithm": "fedavg"},
        "training": {"batch_size": 32, "lr": 0.05},
        "data": {"adapter": "synthetic",
                 "params": {"num_samples": 256, "num_features": 16, "num_classes": 4}},
        "model": {"factory": "models.mlp",
                  "params": {"in_features": 16, "hidden": 32, "num_classes": 4}},
        "backend": "rccl",
    })
    loop = FLRoundLoop(cfg, 0, 1, torch.device("cuda:0"))
    pending = None
    metrics = []
    for r in range(4):
        loop.run_round(r)
        if pending is not None:
            metrics.append(loop.metrics_from(pending))
        pending = loop.evaluate_round_async(r)
    metrics.append(loop.metrics_from(pending))
    assert [m["round"] for m in metrics] == [0, 1, 2, 3]
    assert metrics[-1]["accuracy"] > metrics[0]["accuracy"] - 0.2  # learning-ish


def test_async_eval_pipelined_matches_sync_eval():
    """Multi-stream stress: the side-stream evaluator pipelined
    with next-round training must produce the SAME metrics as synchronous
    evaluation of the same states — catches snapshot/stream races (e.g. the
    round-1 missing reverse sync) deterministically."""
    import os

    import torch.distributed as dist

    from murmura_amd.config.schema import Config
    from murmura_amd.parallel.node_process import FLRoundLoop, init_distributed

    cfg = Config(**{
        "experiment": {"name": "race", "seed": 7, "rounds": 6, "verbose": False},
        "topology": {"type": "fully", "num_nodes": 1},
        "aggregation": {"algorithm": "fedavg"},
        "training": {"local_epochs": 1, "batch_size": 32, "lr": 0.05},
        "data": {"adapter": "synthetic",
                 "params": {"num_samples": 256, "num_features": 20,
                            "num_classes": 4}},
        "model": {"factory": "models.mlp",
                  "params": {"in_features": 20, "hidden": 32, "num_classes": 4}},
        "compute": {"dtype": "fp32"},
    })
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29733")
    if not dist.is_initialized():
        dist.init_process_group("nccl" if torch.cuda.is_available() else "gloo",
                                rank=0, world_size=1)
    device = torch.device("cuda:0")
    torch.cuda.set_device(device)

    # run 1: pipelined (async side-stream eval overlapping next training)
    loop = FLRoundLoop(cfg, 0, 1, device)
    pend, async_rows = None, []
    for r in range(6):
        loop.run_round(r)
        if pend is not None:
            async_rows.append(loop.metrics_from(pend))
        pend = loop.evaluate_round_async(r)
    async_rows.append(loop.metrics_from(pend))

    # run 2: fully synchronous eval after each round (fresh identical loop)
    loop2 = FLRoundLoop(cfg, 0, 1, device)
    sync_rows = []
    for r in range(6):
        loop2.run_round(r)
        sync_rows.append(loop2.evaluate_round(r))

    for a, s in zip(async_rows, sync_rows):
        assert abs(a["accuracy"] - s["accuracy"]) < 1e-5, (a, s)
        assert abs(a["loss"] - s["loss"]) < 1e-4, (a, s)
