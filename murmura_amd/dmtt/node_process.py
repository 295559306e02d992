"""DMTT distributed round loop (reference: murmura/dmtt/node_process.py:53-406).

Extends the base FL round: per-round collaborators C_i^t are the TopB-scored
subset of G^t direct neighbors selected at the END of the previous round
(round 0 uses G^0 neighbors); each round exchanges MODEL_STATE (flat buffer
over RCCL P2P) plus a TOPO_CLAIM (tiny host-side all-gather — the reference
sent a ~tens-of-bytes pickled dict, SURVEY.md §2.7 call-site 6).

RCCL-specific design: C_i^t is ASYMMETRIC (i may select j while j doesn't
select i) but P2P needs both ends to post matching ops, so the exchange set is
symmetrized by one all-gather of the N-bit want-mask per round
(SURVEY.md §7 hard-part 2). Intra-box RCCL delivery is reliable, so the
link-reliability EMA sees ack=1 from actual delivery; the EMA machinery is
kept for the multi-machine/lossy case and for parity with the protocol.
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from murmura_amd.attacks.topology_liar import TopologyLiarAttack
from murmura_amd.config.schema import Config
from murmura_amd.dmtt.state import DMTTNodeState
from murmura_amd.parallel import exchange
from murmura_amd.parallel.node_process import FLRoundLoop


class DMTTRoundLoop(FLRoundLoop):
    def __init__(self, config: Config, rank: int, world_size: int, device: torch.device):
        super().__init__(config, rank, world_size, device)
        if config.dmtt is None:
            raise ValueError("DMTTRoundLoop requires config.dmtt")
        if self.mobility is None:
            raise ValueError("DMTT requires a mobility model (dynamic G^t)")
        self.dmtt_state = DMTTNodeState(rank, world_size, config.dmtt)
        self._collaborators: Optional[List[int]] = None  # C_i^t from previous round

    # ------------------------------------------------------------------
    def _claim_for(self, true_neighbors: List[int], round_num: int) -> List[int]:
        if (
            isinstance(self.attack, TopologyLiarAttack)
            and self.attack.is_compromised(self.rank)
        ):
            return self.attack.get_false_claims(self.rank, true_neighbors, round_num)
        return list(true_neighbors)

    def _score_neighbor_models(self, received: Dict[int, torch.Tensor]) -> None:
        """Load each received flat state into the scratch model and score
        accuracy + vacuity on local data (reference: dmtt/node_process.py:309-363
        did a fresh model + deepcopy per neighbor; here it is one flat copy).

        All candidates are scored with the metrics left ON DEVICE, then read
        back in ONE host sync (the reference's per-neighbor ``float()`` pulls
        forced m syncs per round)."""
        ctx = self.node._get_eval_context()
        if ctx is None:
            return
        order = list(received.keys())
        if not order:
            return
        vac, acc = ctx.evidential_scores(
            torch.stack([received[j] for j in order]), max_samples=100
        )
        vals = torch.stack([vac, acc]).tolist()  # single host sync
        for idx, j in enumerate(order):
            self.dmtt_state.record_model_score(j, vals[0][idx], vals[1][idx])

    def _process_topo_claims(
        self, claims: Dict[int, List[int]], topo, round_num: int
    ) -> None:
        """Verify each peer's claimed neighborhood against the locally
        computed deterministic G^t -> Beta evidence update
        (reference: dmtt/node_process.py:369-395)."""
        n = self.world
        for j, claimed in claims.items():
            if j == self.rank:
                continue
            true_set = set(topo.neighbors[j])
            claimed_set = set(claimed)
            matches = len(true_set & claimed_set)
            contradictions = len(claimed_set - true_set) + len(true_set - claimed_set)
            self.dmtt_state.update_topology_evidence(
                j, d=float(matches), c=0.0, x=float(contradictions)
            )

    # ------------------------------------------------------------------
    def run_round(self, round_num: int) -> None:
        cfg = self.config
        topo = self.topology_at(round_num)
        g_neighbors = list(topo.neighbors[self.rank])

        # collaborators for this round: previous TopB, or G^0 neighbors
        if self._collaborators is None:
            collaborators = g_neighbors
        else:
            collaborators = self._collaborators

        # 1. local train (honest)
        if not self._is_compromised(self.rank):
            self.node.local_train(
                epochs=cfg.training.local_epochs, lr=cfg.training.lr, round_num=round_num
            )
        # 2. snapshot; attack alters only the broadcast copy (own aggregation
        # keeps the clean snapshot, reference: node.py:234)
        own = self.node.get_state()
        wire = own
        if self._is_compromised(self.rank):
            wire = self.attack.apply_attack(self.rank, own, round_num)

        # 3a. symmetrize the asymmetric collaborator sets -> exchange plan
        sym_sets = exchange.symmetrize_wants(collaborators, self.world)
        peers = sym_sets[self.rank]
        received = exchange.exchange_with_neighbors(wire, peers)

        # 3b. topology claims: tiny host-side all-gather (host group — under
        # nccl an object collective would otherwise run on the device)
        claim = self._claim_for(g_neighbors, round_num)
        all_claims: List[Optional[List[int]]] = [None] * self.world
        dist.all_gather_object(all_claims, claim, group=exchange.host_group())
        claims = {j: all_claims[j] for j in range(self.world) if all_claims[j] is not None}

        # 4. trust updates
        for j in peers:
            self.dmtt_state.update_link_reliability(j, ack=j in received)
        self._score_neighbor_models(received)
        self._process_topo_claims(claims, topo, round_num)

        # 5. aggregate with what we received from OUR selected collaborators
        use = [j for j in collaborators if j in received] or peers
        if use:
            stacked = torch.stack([received[j] for j in use], dim=0)
        else:
            stacked = own.new_zeros((0, own.numel()))
        new_state = self.node.aggregate_with_neighbors(
            own, stacked, neighbor_ids=use, round_num=round_num
        )
        self.node.set_state(new_state)

        # 6. select next round's collaborators from G^t direct neighbors
        self._collaborators = self.dmtt_state.top_b(g_neighbors)
        if os.environ.get("MURMURA_DMTT_DEBUG") == "1" and self.rank == 0:
            qs = {j: round(self.dmtt_state.collaborator_score(j), 3)
                  for j in g_neighbors}
            comp = ([j for j in range(self.world)
                     if self.attack is not None and self.attack.is_compromised(j)])
            print(f"[dmtt r{round_num}] rank0 nbrs={g_neighbors} q={qs} "
                  f"chose={self._collaborators} used={use} compromised={comp}",
                  flush=True)
