"""Checkpoint / resume.

The reference persists nothing; with flat state buffers a
checkpoint is one contiguous tensor per node plus the non-float buffers and
the history dict, so this is an MI355X-build extension, not a port.

Format (torch.save):
  {"round": int, "history": dict,
   "nodes": {node_id: {"flat": cpu tensor [P],
                        "nonfloat": {name: tensor},
                        "dtype": str}}}
"""

from __future__ import annotations

from pathlib import Path
from typing import Optional

import torch


def node_state_payload(node) -> dict:
    nonfloat = {
        name: b.detach().cpu().clone()
        for name, b in node.model.named_buffers()
        if not torch.is_floating_point(b)
    }
    return {
        "flat": node.store.flat.detach().float().cpu().clone(),
        "nonfloat": nonfloat,
        "dtype": str(node.dtype),
    }


def restore_node_state(node, payload: dict) -> None:
    node.store.copy_from_flat(payload["flat"].to(node.device, node.dtype))
    buffers = dict(node.model.named_buffers())
    for name, val in payload.get("nonfloat", {}).items():
        if name in buffers:
            buffers[name].copy_(val.to(buffers[name].device))


def save_checkpoint(path, round_num: int, nodes, history: Optional[dict] = None) -> None:
    path = Path(path)
    path.parent.mkdir(parents=True, exist_ok=True)
    blob = {
        "round": round_num,
        "history": history or {},
        "nodes": {n.node_id: node_state_payload(n) for n in nodes},
    }
    tmp = path.with_suffix(path.suffix + ".tmp")
    torch.save(blob, tmp)
    tmp.replace(path)  # atomic


def load_checkpoint(path) -> dict:
    return torch.load(Path(path), map_location="cpu", weights_only=False)


def restore_network(network, blob: dict) -> int:
    """Restore all node states + history; returns the next round index."""
    for node in network.nodes:
        if node.node_id in blob["nodes"]:
            restore_node_state(node, blob["nodes"][node.node_id])
    if blob.get("history"):
        network.history = blob["history"]
    return int(blob["round"]) + 1


def save_rank_checkpoint(path, round_num: int, node, history=None) -> None:
    """Per-rank checkpoint for the distributed backend (each rank persists its
    own node; rank 0 additionally persists the history)."""
    save_checkpoint(path, round_num, [node], history)


def restore_rank(node, blob: dict) -> int:
    if node.node_id in blob["nodes"]:
        restore_node_state(node, blob["nodes"][node.node_id])
    return int(blob["round"]) + 1
