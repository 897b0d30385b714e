"""Static-shape batch padding for hipGraph-captured training.

hipGraph replay (see ``hydragnn_amd/train/captured.py``) requires every
batch to have identical tensor shapes.  Graph batches naturally vary in
node/edge count, so this module pads each collated ``Batch`` to fixed
capacities by appending ONE sentinel "pad graph":

- pad nodes are placed on a line with spacing ``pad_spacing`` (choose
  > the model's interaction cutoff) so every pad edge is longer than
  the cutoff;
- pad edges connect only pad nodes (never a real node), so real-graph
  predictions are bitwise unaffected;
- ``loss_weight_g`` carries per-graph loss weights with the pad graph
  at 0, and the loss functions honour it (weighted mean), so the
  padded loss/gradients equal the unpadded ones exactly.

This replaces the role of the reference's dynamic PyG batching for the
capture fast path (no reference equivalent: HydraGNN re-launches eager
kernels per batch; on MI355X we replay one hipGraph instead).
"""

from __future__ import annotations

import math
from typing import Sequence

import torch

from ..data import Batch, Data, _EDGE_KEYS, _GRAPH_KEYS, _NODE_KEYS

# Keys that index elements (must stay valid): padded with 1, not 0.
_INDEX_LIKE_NODE_KEYS = {"z", "atomic_numbers"}
# Bookkeeping keys handled explicitly.
_SPECIAL = {
    "edge_index", "batch", "ptr", "num_graphs_", "num_nodes_",
    "edge_counts_", "edges_sorted_", "loss_weight_g",
    "num_real_graphs_", "static_shape_",
}


def _pad_rows(t: torch.Tensor, n: int, value: float = 0.0) -> torch.Tensor:
    pad_shape = (n,) + tuple(t.shape[1:])
    pad = t.new_full(pad_shape, value)
    return torch.cat([t, pad], dim=0)


def pad_batch_static(
    batch: Batch,
    node_cap: int,
    edge_cap: int,
    pad_spacing: float = 30.0,
) -> Batch:
    """Pad ``batch`` in place to exactly ``node_cap`` nodes and
    ``edge_cap`` edges by appending one sentinel pad graph.

    Requires ``node_cap >= num_nodes + 2`` (the pad graph needs at
    least 2 nodes to host pad edges).
    """
    if batch.get("y_loc") is not None:
        raise NotImplementedError(
            "static-shape padding does not support y_loc multihead "
            "y-packing yet (the pad graph would shift the per-sample "
            "offsets); use it with MLIP / single-head targets")
    n = batch.num_nodes
    e = batch.num_edges
    b = batch.num_graphs
    n_pad = node_cap - n
    e_pad = edge_cap - e
    if n_pad < 2:
        raise ValueError(
            f"node_cap={node_cap} must exceed batch nodes {n} by >= 2")
    if e_pad < 0:
        raise ValueError(
            f"edge_cap={edge_cap} < batch edges {e}")

    device = batch.batch.device
    ei = batch.edge_index

    # Pad edges: dst non-decreasing (keeps edges_sorted_), src = next
    # pad node on the line -> every pad edge has length >= pad_spacing.
    if e_pad > 0:
        idx = torch.arange(e_pad, device=device)
        dst_local = (idx * n_pad) // e_pad          # non-decreasing
        src_local = (dst_local + 1) % n_pad
        pad_ei = torch.stack([src_local + n, dst_local + n], dim=0)
        batch["edge_index"] = torch.cat([ei, pad_ei], dim=1)
    first = batch  # key-role detection uses the batch itself

    for key in list(batch.keys()):
        if key in _SPECIAL:
            continue
        v = batch.get(key)
        if not torch.is_tensor(v):
            continue
        if key in _INDEX_LIKE_NODE_KEYS:
            batch[key] = _pad_rows(v, n_pad, 1)
        elif key == "pos":
            # 3D grid with spacing pad_spacing: every pad-pair
            # distance >= pad_spacing (> cutoff), and coordinates stay
            # small enough that bf16 rounding cannot collapse two pad
            # nodes onto each other (a zero-length edge would NaN the
            # spherical harmonics and poison the masked loss).
            base = pad_spacing * 10.0
            k = max(2, int(math.ceil(n_pad ** (1.0 / 3.0))))
            idx = torch.arange(n_pad, device=device)
            pad = v.new_zeros(n_pad, v.shape[1])
            pad[:, 0] = base + pad_spacing * (idx % k).to(v.dtype)
            if v.shape[1] > 1:
                pad[:, 1] = base + pad_spacing * ((idx // k) % k).to(
                    v.dtype)
            if v.shape[1] > 2:
                pad[:, 2] = base + pad_spacing * (idx // (k * k)).to(
                    v.dtype)
            batch[key] = torch.cat([v, pad], dim=0)
        elif key in _NODE_KEYS or (
                v.dim() > 0 and v.shape[0] == n and key not in _GRAPH_KEYS
                and key not in _EDGE_KEYS):
            batch[key] = _pad_rows(v, n_pad)
        elif key in _EDGE_KEYS or (v.dim() > 0 and v.shape[0] == e
                                   and key.startswith("edge")):
            batch[key] = _pad_rows(v, e_pad)
        elif key in _GRAPH_KEYS or (v.dim() > 0 and v.shape[0] == b):
            batch[key] = _pad_rows(v, 1)
        # scalars / unrecognized: leave untouched

    w = batch.get("loss_weight_g")
    if w is None:
        w = torch.ones(b, device=device)
    w = torch.cat([w.reshape(-1).float(),
                   torch.zeros(1, device=device)])
    batch["loss_weight_g"] = w

    batch["batch"] = torch.cat([
        batch.batch, torch.full((n_pad,), b, dtype=batch.batch.dtype,
                                device=device)])
    batch["ptr"] = torch.cat([
        batch.ptr, torch.tensor([node_cap], dtype=batch.ptr.dtype,
                                device=device)])
    if "edge_counts_" in batch:
        batch["edge_counts_"] = torch.cat([
            batch["edge_counts_"],
            torch.tensor([e_pad], dtype=batch["edge_counts_"].dtype,
                         device=device)])
    batch["num_graphs_"] = b + 1
    batch["num_real_graphs_"] = b
    batch.num_nodes = node_cap
    batch["static_shape_"] = True
    return batch


class StaticShapeCollater:
    """DataLoader ``collate_fn`` producing fixed-shape padded batches.

    ``node_cap``/``edge_cap`` must cover the largest batch the sampler
    can produce (use :func:`compute_static_caps`).
    """

    def __init__(self, node_cap: int, edge_cap: int,
                 pad_spacing: float = 30.0):
        self.node_cap = int(node_cap)
        self.edge_cap = int(edge_cap)
        self.pad_spacing = float(pad_spacing)

    def __call__(self, data_list: Sequence[Data]) -> Batch:
        b = Batch.from_data_list(list(data_list))
        return pad_batch_static(b, self.node_cap, self.edge_cap,
                                self.pad_spacing)


def compute_static_caps(dataset, batch_size: int,
                        node_margin: int = 2,
                        edge_margin: int = 0,
                        sequential: bool = False,
                        max_pad_edges_per_node: int = 64):
    """(node, edge) capacities for batches of ``batch_size`` samples.

    ``sequential=True``: exact max over consecutive batches (loaders
    with shuffle=False) — tightest caps.  ``sequential=False``: worst
    case (the batch_size largest per-sample counts summed) — safe for
    shuffled samplers.

    The node cap is widened so pad edges spread over enough pad nodes
    that no pad node receives more than ``max_pad_edges_per_node``
    edges: a batch at the minimum edge count pads edge_cap-E edges,
    and concentrating them on O(1) pad nodes creates giant scatter
    segments that serialize the segment-reduce kernels (measured 10x
    step blowup on MI355X)."""
    n_counts = [d.num_nodes for d in dataset]
    e_counts = [d.num_edges for d in dataset]
    if sequential:
        node_sums = [sum(n_counts[i:i + batch_size])
                     for i in range(0, len(n_counts), batch_size)]
        edge_sums = [sum(e_counts[i:i + batch_size])
                     for i in range(0, len(e_counts), batch_size)]
        # ignore a trailing short batch for the min (drop_last loaders)
        full = [s for s, c in zip(
            edge_sums, range(0, len(e_counts), batch_size))
            if c + batch_size <= len(e_counts)]
        node_cap = max(node_sums) + max(node_margin, 2)
        edge_cap = max(edge_sums) + edge_margin
        min_edges = min(full) if full else min(edge_sums)
    else:
        nodes = sorted(n_counts, reverse=True)
        edges = sorted(e_counts, reverse=True)
        node_cap = sum(nodes[:batch_size]) + max(node_margin, 2)
        edge_cap = sum(edges[:batch_size]) + edge_margin
        min_edges = sum(sorted(e_counts)[:batch_size])
    max_pad_e = max(edge_cap - min_edges, 0)
    extra_nodes = min(-(-max_pad_e // max_pad_edges_per_node), 8192)
    node_cap += extra_nodes
    return node_cap, edge_cap
