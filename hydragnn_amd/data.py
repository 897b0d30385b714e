"""Graph data containers for hydragnn_amd.

MI355X-native replacement for the torch_geometric ``Data``/``Batch``
containers the reference framework builds on (HydraGNN uses PyG Data
everywhere, e.g. /root/reference/hydragnn/preprocess/load_data.py).
We keep the same attribute conventions (``x``, ``pos``, ``edge_index``,
``edge_attr``, ``y``, ``batch``) so configs and user code translate
directly, but the implementation is self-contained: no PyG dependency,
collation is a single pass with pinned-memory-friendly contiguous
tensors, and collation records whether edges are destination-sorted
(``edges_sorted_``) so the deterministic CSR segment-reduction kernels
can be used without re-sorting per step.
"""

from __future__ import annotations

import copy
from typing import Any, Dict, Optional, Sequence

import torch

# Attributes indexed per-node (concatenate along dim 0, no offset).
_NODE_KEYS = {
    "x", "pos", "forces", "vel", "node_attrs", "z", "atomic_numbers",
    "y_node",
}
# Attributes indexed per-edge.
_EDGE_KEYS = {"edge_attr", "edge_shifts", "edge_lengths"}
# Attributes indexed per-graph (stack / cat along dim 0).
_GRAPH_KEYS = {
    "y", "energy", "cell", "pbc", "dataset_name", "graph_attr", "stress",
    "supercell_size",
}


class Data:
    """A single graph sample.

    Tensors of interest:
      x           [N, F]   node features
      pos         [N, 3]   node positions
      edge_index  [2, E]   COO connectivity (row 0 = src, row 1 = dst)
      edge_attr   [E, Fe]  edge features
      edge_shifts [E, 3]   PBC shift vectors (S @ cell), optional
      y           [...]    targets
    """

    def __init__(self, **kwargs: Any) -> None:
        self._store: Dict[str, Any] = {}
        for k, v in kwargs.items():
            setattr(self, k, v)

    # -- attribute plumbing -------------------------------------------------
    def __getattr__(self, key: str) -> Any:
        store = object.__getattribute__(self, "__dict__").get("_store")
        if store is not None and key in store:
            return store[key]
        raise AttributeError(
            f"'{self.__class__.__name__}' object has no attribute '{key}'"
        )

    def __setattr__(self, key: str, value: Any) -> None:
        if key.startswith("_"):
            object.__setattr__(self, key, value)
        else:
            self._store[key] = value

    def __delattr__(self, key: str) -> None:
        if key in self._store:
            del self._store[key]
        else:
            object.__delattr__(self, key)

    def __contains__(self, key: str) -> bool:
        return key in self._store

    def __getitem__(self, key: str) -> Any:
        return self._store[key]

    def __setitem__(self, key: str, value: Any) -> None:
        self._store[key] = value

    def __delitem__(self, key: str) -> None:
        del self._store[key]

    def get(self, key: str, default: Any = None) -> Any:
        return self._store.get(key, default)

    def keys(self):
        return self._store.keys()

    def items(self):
        return self._store.items()

    def to_dict(self) -> Dict[str, Any]:
        return dict(self._store)

    @classmethod
    def from_dict(cls, mapping: Dict[str, Any]) -> "Data":
        out = cls()
        for k, v in mapping.items():
            out[k] = v
        return out

    @classmethod
    def from_pyg(cls, data) -> "Data":
        """Convert a torch_geometric ``Data``-like object (anything
        exposing per-key tensor attributes via ``keys``/attribute
        access — duck-typed, no PyG import) so reference-era datasets
        drop straight into this framework's loaders."""
        out = cls()
        keys = data.keys() if callable(getattr(data, "keys", None)) \
            else getattr(data, "keys", [])
        for k in list(keys):
            v = data[k] if hasattr(data, "__getitem__") \
                else getattr(data, k)
            out[k] = v
        n = getattr(data, "num_nodes", None)
        if n is not None:
            out.num_nodes = int(n)
        return out

    def clone(self) -> "Data":
        out = self.__class__()
        for k, v in self._store.items():
            out._store[k] = v.clone() if torch.is_tensor(v) else copy.deepcopy(v)
        return out

    # -- shape helpers ------------------------------------------------------
    @property
    def num_nodes(self) -> int:
        if "num_nodes_" in self._store:
            return int(self._store["num_nodes_"])
        for key in ("x", "pos", "z"):
            v = self._store.get(key)
            if v is not None:
                return v.shape[0]
        ei = self._store.get("edge_index")
        if ei is not None and ei.numel() > 0:
            return int(ei.max()) + 1
        return 0

    @num_nodes.setter
    def num_nodes(self, value: int) -> None:
        self._store["num_nodes_"] = int(value)

    @property
    def num_edges(self) -> int:
        ei = self._store.get("edge_index")
        return 0 if ei is None else ei.shape[1]

    def __repr__(self) -> str:
        parts = []
        for k, v in self._store.items():
            if torch.is_tensor(v):
                parts.append(f"{k}={list(v.shape)}")
            else:
                parts.append(f"{k}={v!r}")
        return f"{self.__class__.__name__}({', '.join(parts)})"

    # -- device movement ----------------------------------------------------
    def to(self, device, non_blocking: bool = False) -> "Data":
        for k, v in self._store.items():
            if torch.is_tensor(v):
                self._store[k] = v.to(device, non_blocking=non_blocking)
        return self

    def cpu(self) -> "Data":
        return self.to("cpu")

    def pin_memory(self) -> "Data":
        for k, v in self._store.items():
            if torch.is_tensor(v) and v.device.type == "cpu":
                self._store[k] = v.pin_memory()
        return self


def _is_node_key(key: str, data: Data, value: Any) -> bool:
    if key in _NODE_KEYS:
        return True
    if key in _EDGE_KEYS or key in _GRAPH_KEYS:
        return False
    if torch.is_tensor(value) and value.dim() > 0:
        n = data.num_nodes
        if n > 0 and value.shape[0] == n and key not in ("edge_index",):
            # Heuristic consistent with PyG: leading dim == num_nodes.
            e = data.num_edges
            if value.shape[0] == e and key.startswith("edge"):
                return False
            return True
    return False


def _is_edge_key(key: str, data: Data, value: Any) -> bool:
    if key in _EDGE_KEYS:
        return True
    if key in _NODE_KEYS or key in _GRAPH_KEYS:
        return False
    if torch.is_tensor(value) and value.dim() > 0:
        e = data.num_edges
        if e > 0 and value.shape[0] == e and key.startswith("edge"):
            return True
    return False


class Batch(Data):
    """A batch of graphs collated into one big disconnected graph.

    Adds:
      batch          [N]    graph id per node
      ptr            [B+1]  node offsets per graph
      edges_sorted_  bool   whether edge_index is dst-sorted (enables
                            the CSR segment-reduce fast path)
    """

    @classmethod
    def from_data_list(cls, data_list: Sequence[Data]) -> "Batch":
        assert len(data_list) > 0
        batch = cls()
        keys = list(data_list[0].keys())
        node_counts = [d.num_nodes for d in data_list]
        edge_counts = [d.num_edges for d in data_list]
        n_total = sum(node_counts)

        device = None
        for d in data_list:
            for v in d._store.values():
                if torch.is_tensor(v):
                    device = v.device
                    break
            if device is not None:
                break

        ptr = torch.zeros(len(data_list) + 1, dtype=torch.long, device=device)
        ptr[1:] = torch.as_tensor(node_counts, device=device).cumsum(0)
        batch_vec = torch.repeat_interleave(
            torch.arange(len(data_list), device=device),
            torch.as_tensor(node_counts, device=device),
        )

        first = data_list[0]
        for key in keys:
            v0 = first.get(key)
            if key == "num_nodes_":
                continue
            if key == "edge_index":
                parts = [d.edge_index for d in data_list]
                if parts:
                    # one vectorized offset add instead of B tiny adds
                    # (collating 8k-graph batches was loader-bound)
                    ei = torch.cat(parts, dim=1)
                    off = torch.repeat_interleave(
                        ptr[:-1],
                        torch.as_tensor(edge_counts, device=device))
                    batch["edge_index"] = ei + off.unsqueeze(0)
                else:
                    batch["edge_index"] = torch.zeros(
                        2, 0, dtype=torch.long)
                continue
            if not torch.is_tensor(v0):
                batch[key] = [d.get(key) for d in data_list]
                continue
            if _is_node_key(key, first, v0) or _is_edge_key(key, first, v0):
                batch[key] = torch.cat([d.get(key) for d in data_list], dim=0)
            else:
                vals = [d.get(key) for d in data_list]
                if v0.dim() == 0:
                    batch[key] = torch.stack(vals, dim=0)
                else:
                    batch[key] = torch.cat(vals, dim=0)

        batch["batch"] = batch_vec
        batch["ptr"] = ptr
        batch.num_nodes = n_total
        batch["num_graphs_"] = len(data_list)
        ei = batch.get("edge_index")
        if ei is not None and ei.numel() > 0 and not ei.is_cuda:
            dst = ei[1]
            batch["edges_sorted_"] = bool(
                (dst[1:] >= dst[:-1]).all())
        batch["edge_counts_"] = torch.as_tensor(edge_counts, device=device)
        return batch

    @property
    def num_graphs(self) -> int:
        if "num_graphs_" in self._store:
            return int(self._store["num_graphs_"])
        if "ptr" in self._store:
            return self._store["ptr"].numel() - 1
        if "batch" in self._store and self._store["batch"].numel() > 0:
            return int(self._store["batch"].max()) + 1
        return 1

    def sort_edges_by_dst(self) -> "Batch":
        """Sort edge_index (and aligned edge attrs) by destination node and
        attach a CSR rowptr. Called once per batch so the HIP segment
        kernels can run a row-per-wave reduction instead of atomics."""
        ei = self.get("edge_index")
        if ei is None or ei.numel() == 0:
            self["rowptr"] = torch.zeros(
                self.num_nodes + 1, dtype=torch.long,
                device=None if ei is None else ei.device)
            return self
        dst = ei[1]
        perm = torch.argsort(dst, stable=True)
        self["edge_index"] = ei[:, perm]
        for key in list(self._store.keys()):
            if key == "edge_index":
                continue
            v = self._store[key]
            if torch.is_tensor(v) and _is_edge_key(key, self, v):
                self._store[key] = v[perm]
        counts = torch.bincount(self["edge_index"][1], minlength=self.num_nodes)
        rowptr = torch.zeros(
            self.num_nodes + 1, dtype=torch.long, device=ei.device)
        rowptr[1:] = counts.cumsum(0)
        self["rowptr"] = rowptr
        return self


def to_dense_batch(
    x: torch.Tensor,
    batch: Optional[torch.Tensor],
    fill_value: float = 0.0,
    max_num_nodes: Optional[int] = None,
    batch_size: Optional[int] = None,
):
    """Pad per-graph node features into [B, maxN, F] + mask [B, maxN].

    Same contract as torch_geometric.utils.to_dense_batch (used by the
    reference GPS layer, /root/reference/hydragnn/globalAtt/gps.py:178).
    """
    if batch is None:
        batch = torch.zeros(x.shape[0], dtype=torch.long, device=x.device)
    if batch_size is None:
        batch_size = int(batch.max()) + 1 if batch.numel() > 0 else 1
    counts = torch.bincount(batch, minlength=batch_size)
    if max_num_nodes is None:
        max_num_nodes = int(counts.max()) if counts.numel() > 0 else 0
    ptr = torch.zeros(batch_size + 1, dtype=torch.long, device=x.device)
    ptr[1:] = counts.cumsum(0)
    idx_in_graph = torch.arange(x.shape[0], device=x.device) - ptr[batch]
    valid = idx_in_graph < max_num_nodes
    flat_idx = batch * max_num_nodes + idx_in_graph
    out = x.new_full((batch_size * max_num_nodes,) + x.shape[1:], fill_value)
    out[flat_idx[valid]] = x[valid]
    out = out.view(batch_size, max_num_nodes, *x.shape[1:])
    mask = torch.zeros(
        batch_size * max_num_nodes, dtype=torch.bool, device=x.device)
    mask[flat_idx[valid]] = True
    mask = mask.view(batch_size, max_num_nodes)
    return out, mask
