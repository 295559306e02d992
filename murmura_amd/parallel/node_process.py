"""Per-rank FL node loop over torch.distributed — one process per MI355X GPU.

Replaces the reference's ZMQ NodeProcess (murmura/distributed/node_process.py:60-364):
- rank == node_id; device = cuda:LOCAL_RANK (RCCL over xGMI) or cpu (gloo, tests)
- rounds are synchronized by the collectives themselves (deterministic
  intra-box), not by wall-clock windows; an optional round_duration_s budget is
  kept only for parity experiments
- per-node seeding seed + node_id (reference: node_process.py:113)
- the per-round topology (static or mobility G^t) is computed locally and
  identically on every rank from the shared seed — no coordination messages

Round structure (mirrors Network.train semantics exactly so histories match
the simulation backend on the same seeds):
  1. local train (honest nodes)
  2. snapshot flat state; compromised ranks attack only the BROADCAST copy
     (their own aggregation uses the clean snapshot, reference: node.py:234
     own_state = self.get_state(); the attack alters what neighbors receive,
     network.py:110-119)
  3. exchange: all-reduce fast path (fedavg + fully-connected) or grouped
     RCCL P2P along this round's edges
  4. aggregate (HIP kernels) and apply
  5. evaluate + gather metrics to rank 0 (host-side, tiny)
"""

from __future__ import annotations

import datetime
import os
from typing import Dict, List, Optional

import torch
import torch.distributed as dist
from torch.utils.data import DataLoader

from murmura_amd.config.schema import Config
from murmura_amd.core.network import new_history
from murmura_amd.core.node import Node
from murmura_amd.parallel import exchange
from murmura_amd.topology.generators import create_topology
from murmura_amd.utils import factories
from murmura_amd.utils.seed import set_seed
from murmura_amd.utils.timing import PhaseTimer


def _resolve_backend(config: Config) -> str:
    env = os.environ.get("MURMURA_COMM_BACKEND")
    if env:
        return env
    b = config.distributed.comm_backend
    if b != "auto":
        return b
    return "nccl" if torch.cuda.is_available() else "gloo"


def init_distributed(config: Config, rank: int, world_size: int) -> torch.device:
    """Initialize the process group; returns this rank's device."""
    backend = _resolve_backend(config)
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", config.distributed.master_addr)
        os.environ.setdefault("MASTER_PORT", str(config.distributed.master_port))
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world_size,
            timeout=datetime.timedelta(seconds=300),
        )
    if backend == "nccl" or (backend == "gloo" and torch.cuda.is_available()
                              and os.environ.get("MURMURA_GLOO_CUDA") == "1"):
        ndev = max(1, torch.cuda.device_count())
        local = int(os.environ.get("LOCAL_RANK", rank)) % ndev
        device = torch.device(f"cuda:{local}")
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
    # gloo subgroup for host-side object/metadata collectives (collective —
    # every rank calls it here; see exchange.host_group)
    exchange.create_host_group_if_needed()
    return device


def build_node(config: Config, rank: int, device: torch.device) -> Node:
    """Construct this rank's Node — mirrors Network.from_config per-node setup
    (reference re-loads data per process too, node_process.py:333-364)."""
    set_seed(config.experiment.seed + rank,
             deterministic_kernels=config.compute.deterministic_kernels)
    model_factory = factories.build_model_factory(config)
    adapter = factories.build_dataset_adapter(config)
    agg_factory = factories.build_aggregator_factory(config, model_factory)
    criterion_factory = factories.build_criterion_factory(config)
    evidential = factories.is_evidential(config)
    dtype = factories.compute_dtype(config)

    client_data = adapter.get_client_data(rank)
    n_samples = len(client_data)
    bs = min(config.training.batch_size, max(2, n_samples))
    g = torch.Generator().manual_seed(config.experiment.seed + rank)
    train_loader = DataLoader(
        client_data, batch_size=bs, shuffle=True, drop_last=n_samples > bs, generator=g
    )
    test_loader = DataLoader(client_data, batch_size=bs, shuffle=False)
    # per-node deterministic init: identical weights as the simulation backend
    with torch.random.fork_rng(devices=[]):
        torch.manual_seed(config.experiment.seed + rank)
        model = model_factory()
    return Node(
        node_id=rank,
        model=model,
        train_loader=train_loader,
        test_loader=test_loader,
        aggregator=agg_factory(rank),
        device=device,
        criterion=criterion_factory() if criterion_factory else None,
        evidential=evidential,
        dtype=dtype,
        model_factory=model_factory,
        channels_last=config.compute.channels_last,
    )


def _drop_stragglers(topo, ready: List[bool]):
    """Edges restricted to ready nodes (stragglers neither send nor receive
    this round — everyone computes the same reduced edge set locally)."""
    from murmura_amd.topology.base import Topology

    edges = {
        (i, j) for i, j in topo.edges if ready[i] and ready[j]
    }
    return Topology.from_edges(topo.num_nodes, edges)


class FLRoundLoop:
    """The distributed round loop, factored so bench.py can time single rounds."""

    def __init__(self, config: Config, rank: int, world_size: int, device: torch.device):
        if config.topology.num_nodes != world_size:
            raise ValueError(
                f"world_size {world_size} != topology.num_nodes {config.topology.num_nodes}"
            )
        self.config = config
        self.rank = rank
        self.world = world_size
        self.device = device
        self.node = build_node(config, rank, device)
        self.topology = create_topology(
            config.topology.type,
            world_size,
            p=config.topology.p,
            k=config.topology.k,
            seed=config.topology.seed,
        )
        self.mobility = factories.build_mobility_model(config)
        self.attack = factories.build_attack(config)
        self.sketch_mode = config.distributed.sketch_wire_mode
        self.timer = PhaseTimer(device)
        self._fully_connected = all(
            len(self.topology.neighbors[i]) == world_size - 1 for i in range(world_size)
        )

    def topology_at(self, round_num: int):
        if self.mobility is not None:
            return self.mobility.topology_at(round_num)
        return self.topology

    def _is_compromised(self, node_id: int) -> bool:
        return self.attack is not None and self.attack.is_compromised(node_id)

    def _ready_mask(self, train_seconds: float) -> Optional[List[bool]]:
        """Straggler semantics (reference: node_process.py:210-217 skips the
        exchange when training overruns the round budget; peers aggregate with
        whatever arrived). With collectives the equivalent is agreed
        deterministically: one tiny all-gather of a ready bit; overrun nodes
        are dropped from EVERYONE's edge set this round. Disabled when
        round_duration_s == 0 (intra-box deterministic-synchronous mode)."""
        budget = self.config.distributed.round_duration_s
        if budget <= 0:
            return None
        # host-side values in, host-side decision out: one byte per rank over
        # the host group (gloo subgroup under nccl) — no device round trip
        ready = torch.tensor(
            [1 if train_seconds <= budget else 0], dtype=torch.uint8
        )
        out = [torch.zeros_like(ready) for _ in range(self.world)]
        dist.all_gather(out, ready, group=exchange.host_group())
        return [bool(t.item()) for t in out]

    def run_round(self, round_num: int) -> None:
        import time as _time

        cfg = self.config
        topo = self.topology_at(round_num)
        # 1. local training (honest nodes only; compromised stay frozen)
        t_train0 = _time.perf_counter()
        with self.timer.phase("train"):
            if not self._is_compromised(self.rank):
                self.node.local_train(
                    epochs=cfg.training.local_epochs, lr=cfg.training.lr,
                    round_num=round_num,
                )
        train_seconds = _time.perf_counter() - t_train0
        ready = self._ready_mask(train_seconds)
        if ready is not None:
            topo = _drop_stragglers(topo, ready)
            if not ready[self.rank]:
                # overran the budget: skip exchange, keep own state
                return
        # 2. snapshot; the attack alters only the broadcast copy (wire) —
        # own aggregation keeps the clean snapshot (reference: node.py:234)
        with self.timer.phase("snapshot_attack"):
            own = self.node.get_state()
            wire = own
            if self._is_compromised(self.rank):
                wire = self.attack.apply_attack(self.rank, own, round_num)

        nbr_ids = list(topo.neighbors[self.rank])
        use_allreduce = (
            cfg.aggregation.algorithm == "fedavg"
            and self.mobility is None
            and self._fully_connected
            and self.attack is None
            and (ready is None or all(ready))  # a straggler skipped: no collective
        )
        if use_allreduce:
            # K1 folded into the collective: new state = global mean
            with self.timer.phase("exchange"):
                new_state = exchange.allreduce_mean(wire)
            self.node.set_state(new_state)
            return

        if self.sketch_mode and cfg.aggregation.algorithm == "sketchguard":
            self._run_round_sketch_wire(own, wire, nbr_ids, round_num)
            return

        use_chunked = (
            cfg.distributed.overlap_exchange
            and cfg.aggregation.algorithm in ("krum", "balance")
            and len(nbr_ids) > 0
        )
        if use_chunked:
            # 3+4 overlapped: Gram accumulation (the Krum/BALANCE distance
            # input) runs chunk-by-chunk while later chunks are in flight
            with self.timer.phase("exchange"):
                stacked_all, g = exchange.exchange_chunked_with_gram(
                    own, nbr_ids, wire=wire
                )
            with self.timer.phase("aggregate"):
                from murmura_amd import ops as _ops

                d2 = _ops.sq_dists_from_gram(g)
                own_norm = g[0, 0].clamp_min(0).sqrt()
                new_state = self.node.aggregate_with_neighbors(
                    own, stacked_all[1:], neighbor_ids=nbr_ids,
                    round_num=round_num, pairwise_d2=d2, own_norm=own_norm,
                )
                self.node.set_state(new_state)
            return

        # 3. grouped P2P along this round's edges (send the wire copy)
        with self.timer.phase("exchange"):
            received = exchange.exchange_with_neighbors(wire, nbr_ids)
        if nbr_ids:
            stacked = torch.stack([received[j] for j in nbr_ids], dim=0)
        else:
            stacked = own.new_zeros((0, own.numel()))
        # 4. aggregate + apply
        with self.timer.phase("aggregate"):
            new_state = self.node.aggregate_with_neighbors(
                own, stacked, neighbor_ids=nbr_ids, round_num=round_num
            )
            self.node.set_state(new_state)

    def _run_round_sketch_wire(self, own, wire, nbr_ids, round_num: int) -> None:
        """Sketchguard sketch-first exchange (the comm-saving mode the
        reference left latent, sketchguard.py:114-132):

        1. all-gather the 4 KB Count-Sketches (K4 kernel) — N x S floats
        2. filter locally in sketch space (identical threshold math)
        3. move FULL P-vectors over xGMI only along accepted edges
           (plus the closest-neighbor fallback edge)

        Rejected neighbors' states never cross the wire: per round per node
        the P2P volume drops from |N| x P floats to |accepted| x P + N x S.
        """
        import torch as _t

        agg = self.node.aggregator
        own_sketch = agg.get_sketch(own)  # [S] — clean, for local filtering
        # broadcast the sketch of the WIRE copy (what neighbors would see)
        wire_sk = (own_sketch if wire is own else agg.get_sketch(wire)).contiguous()
        if wire_sk.is_cuda and dist.get_backend() == "gloo":
            wire_sk = wire_sk.cpu()
        all_sk = [_t.empty_like(wire_sk) for _ in range(self.world)]
        dist.all_gather(all_sk, wire_sk)
        all_sk = _t.stack(all_sk).to(own.device)  # [N, S]
        if not nbr_ids:
            self.node.set_state(own)
            return
        nbr_sk = all_sk[_t.tensor(nbr_ids, device=own.device)]
        accept = agg.wire_filter(own_sketch, nbr_sk, round_num)
        # fallback: always fetch the sketch-closest neighbor so the
        # min_neighbors fallback (balance.py:133-135 semantics) has its
        # state. The P2P plan needs host integers, so ONE device->host sync
        # is inherent here; accept-mask and closest-index are combined into
        # that single transfer (round-1 review flagged the extra .item()).
        dists = (nbr_sk.float() - own_sketch.float().unsqueeze(0)).norm(dim=1)
        fetch = accept.clone()
        fetch[dists.argmin()] = True
        accept_idx = [i for i, a in enumerate(fetch.tolist()) if a]
        want = [nbr_ids[i] for i in accept_idx]
        self.last_wire_stats = {
            "accepted": len(accept_idx),
            "neighbors": len(nbr_ids),
            "sketch_bytes": int(all_sk.numel() * 4),
            "state_bytes_exchanged": int(len(want) * own.numel() * own.element_size()),
            "state_bytes_saved": int(
                (len(nbr_ids) - len(want)) * own.numel() * own.element_size()
            ),
        }
        sym = exchange.symmetrize_wants(want, self.world, own.device)
        received = exchange.exchange_with_neighbors(wire, sym[self.rank])
        use = [j for j in want if j in received]
        if use:
            stacked = _t.stack([received[j] for j in use], dim=0)
            use_sk = all_sk[_t.tensor(use, device=own.device)]
        else:
            stacked = own.new_zeros((0, own.numel()))
            use_sk = None
        new_state = self.node.aggregate_with_neighbors(
            own,
            stacked,
            neighbor_ids=use,
            round_num=round_num,
            neighbor_sketches=use_sk,
            full_neighbor_count=len(nbr_ids),
        )
        self.node.set_state(new_state)

    def evaluate_round_async(self, round_num: int):
        """Launch evaluation of the current (post-aggregation) state on a side
        stream; returns a handle whose .resolve() yields metric tensors. The
        next round's training overlaps with this evaluation — the evaluator
        snapshots the state first, so there is no race. Falls back to a
        synchronous pseudo-handle on CPU."""
        if self.device.type == "cuda" and self.node.model_factory is not None:
            if not hasattr(self, "_async_eval") or self._async_eval is None:
                from murmura_amd.core.async_eval import AsyncEvaluator

                try:
                    self._async_eval = AsyncEvaluator(self.node)
                except Exception:
                    self._async_eval = False  # unsupported model; stay sync
            if self._async_eval:
                return self._async_eval.launch(round_num)

        class _Sync:
            def __init__(s2, res, rn):
                s2._res, s2.round_num = res, rn

            def resolve(s2):
                return s2._res

        return _Sync(self.node.evaluate(), round_num)

    def metrics_from(self, handle) -> Dict[str, float]:
        res = handle.resolve()
        out = {k: float(v) for k, v in res.items() if isinstance(v, torch.Tensor)}
        out["round"] = handle.round_num
        out["node_id"] = self.rank
        out["compromised"] = self._is_compromised(self.rank)
        return out

    def evaluate_round(self, round_num: int) -> Dict[str, float]:
        with self.timer.phase("evaluate"):
            res = self.node.evaluate()
        out = {k: float(v) for k, v in res.items() if isinstance(v, torch.Tensor)}
        out["round"] = round_num
        out["node_id"] = self.rank
        out["compromised"] = self._is_compromised(self.rank)
        return out


def _append_history(history: Dict[str, List[float]], round_num: int, rows: List[dict]) -> None:
    import math

    accs = [r["accuracy"] for r in rows]
    losses = [r["loss"] for r in rows]
    mean = sum(accs) / len(accs)
    std = math.sqrt(sum((a - mean) ** 2 for a in accs) / len(accs))
    honest = [r["accuracy"] for r in rows if not r["compromised"]]
    comp = [r["accuracy"] for r in rows if r["compromised"]]
    history["round"].append(round_num)
    history["mean_accuracy"].append(mean)
    history["std_accuracy"].append(std)
    history["mean_loss"].append(sum(losses) / len(losses))
    history["honest_accuracy"].append(sum(honest) / len(honest) if honest else 0.0)
    history["compromised_accuracy"].append(sum(comp) / len(comp) if comp else 0.0)
    for key, hkey in [
        ("vacuity", "mean_vacuity"),
        ("entropy", "mean_entropy"),
        ("strength", "mean_strength"),
    ]:
        vals = [r[key] for r in rows if key in r]
        history[hkey].append(sum(vals) / len(vals) if vals else 0.0)


def run_node_process(
    config: Config, rank: int, world_size: int, destroy_group: bool = True,
    checkpoint_dir: Optional[str] = None, checkpoint_every: int = 0,
    resume: bool = False,
) -> Optional[Dict[str, List[float]]]:
    """Full distributed run for one rank; rank 0 returns the history.

    ``checkpoint_dir``: each rank writes node_<rank>.ckpt every
    ``checkpoint_every`` rounds; ``resume`` restores from those files."""
    from murmura_amd.utils import checkpoint as ckpt

    device = init_distributed(config, rank, world_size)
    if config.dmtt is not None:
        from murmura_amd.dmtt.node_process import DMTTRoundLoop

        loop: FLRoundLoop = DMTTRoundLoop(config, rank, world_size, device)
    else:
        loop = FLRoundLoop(config, rank, world_size, device)
    history = new_history() if rank == 0 else None
    start_round = 0
    ckpt_path = None
    if checkpoint_dir is not None:
        import pathlib

        ckpt_path = pathlib.Path(checkpoint_dir) / f"node_{rank}.ckpt"
        if resume and ckpt_path.exists():
            blob = ckpt.load_checkpoint(ckpt_path)
            start_round = ckpt.restore_rank(loop.node, blob)
            if rank == 0 and blob.get("history"):
                history = blob["history"]
    # out-of-band mode: metrics become local file appends (no collective —
    # the reference Monitor's passive/crash-tolerant property, monitor.py:1-15)
    writer = None
    if config.distributed.metrics_dir is not None:
        from murmura_amd.parallel.monitor import MetricsWriter

        writer = MetricsWriter(config.distributed.metrics_dir, rank)

    def _flush(handle) -> None:
        metrics = loop.metrics_from(handle)
        if writer is not None:
            writer.write(metrics)
            return
        rows = exchange.gather_metrics(metrics, dst=0)
        if rank == 0:
            _append_history(history, handle.round_num, rows)
            if config.experiment.verbose:
                print(
                    f"[round {handle.round_num}] "
                    f"acc={history['mean_accuracy'][-1]:.4f} "
                    f"honest={history['honest_accuracy'][-1]:.4f}",
                    flush=True,
                )

    # software-pipelined loop: round r+1's training overlaps round r's
    # side-stream evaluation + metrics gather
    pending = None
    for r in range(start_round, config.experiment.rounds):
        loop.run_round(r)
        # resolve round r-1's eval FIRST: handles share the eval graph's
        # accumulator, so the next launch must not be enqueued before the
        # previous result is read (it already overlapped with this round's
        # training, which is the point)
        if pending is not None:
            _flush(pending)
        pending = loop.evaluate_round_async(r)
        if ckpt_path is not None and checkpoint_every and (r + 1) % checkpoint_every == 0:
            _flush(pending)
            pending = None
            ckpt.save_rank_checkpoint(ckpt_path, r, loop.node,
                                      history if rank == 0 else None)
    if pending is not None:
        _flush(pending)
    if writer is not None:
        writer.close()
        # rank 0 assembles the same history schema from the files so the
        # return contract is identical in both modes
        if rank == 0:
            from murmura_amd.parallel.monitor import assemble_history

            exchange.barrier(device)  # let all ranks finish their appends
            history = assemble_history(config.distributed.metrics_dir, world_size)
        else:
            exchange.barrier(device)
    # aggregate per-node aggregator statistics on rank 0 (the distributed
    # analogue of Network.get_node_statistics)
    stats = loop.node.aggregator.get_statistics()
    stats["phase_timings_ms"] = loop.timer.summary()
    all_stats = exchange.gather_metrics(stats, dst=0)
    if rank == 0 and history is not None:
        history["node_statistics"] = {i: s for i, s in enumerate(all_stats)}
    exchange.barrier(device)
    if destroy_group:
        dist.destroy_process_group()
    return history
