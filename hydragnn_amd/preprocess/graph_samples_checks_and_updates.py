"""Graph-sample checks and update helpers, reference-compatible names
(reference hydragnn/preprocess/graph_samples_checks_and_updates.py).

A user migrating from the reference calls these directly from example
scripts and custom dataset builders; the implementations route onto
this framework's own ops (ops/geometry.py radius graphs on the HIP
pair kernel, preprocess/transforms.py PBC transforms)."""

from __future__ import annotations

from typing import List

import torch

from ..data import Data
from ..ops.geometry import radius_graph, radius_graph_pbc
from ..utils.config.config_utils import (_calculate_avg_deg, _gather_deg,
                                         check_if_graph_size_variable)
from .transforms import pbc_distance, pbc_local_cartesian


class RadiusGraph:
    """Transform: build data.edge_index from positions (open
    boundaries).  Reference graph_samples_checks_and_updates.py:112."""

    def __init__(self, r: float, loop: bool = False,
                 max_num_neighbors: int = 32):
        self.r = r
        self.loop = loop
        self.max_num_neighbors = max_num_neighbors

    def __call__(self, data: Data) -> Data:
        data.edge_index = radius_graph(
            data.pos, self.r, getattr(data, "batch", None),
            max_num_neighbors=self.max_num_neighbors, loop=self.loop)
        return data


class RadiusGraphPBC:
    """Transform: periodic neighbor list with shift vectors (requires
    data.cell and data.pbc)."""

    def __init__(self, r: float, loop: bool = False,
                 max_num_neighbors: int = 1000000):
        self.r = r
        self.loop = loop
        self.max_num_neighbors = max_num_neighbors

    def __call__(self, data: Data) -> Data:
        pbc = getattr(data, "pbc", (True, True, True))
        edge_index, shifts = radius_graph_pbc(
            data.pos, self.r, data.cell, pbc=pbc,
            max_num_neighbors=self.max_num_neighbors, loop=self.loop)
        data.edge_index = edge_index
        data.edge_shifts = shifts
        return data


class RadiusInteractionGraphCPU(RadiusGraph):
    """CPU-forced variant (the radius build already runs on CPU for
    host-resident Data; kept for name parity)."""


class PBCDistance:
    def __init__(self, norm: bool = False, max_length: float = 1.0):
        self.norm = norm
        self.max_length = max_length

    def __call__(self, data: Data) -> Data:
        return pbc_distance(data, norm=self.norm,
                            max_length=self.max_length)


class PBCLocalCartesian:
    def __call__(self, data: Data) -> Data:
        return pbc_local_cartesian(data)


def get_radius_graph(radius, max_neighbours, loop=False):
    return RadiusGraph(r=radius, loop=loop,
                       max_num_neighbors=max_neighbours)


def get_radius_graph_pbc(radius, max_neighbours, loop=False):
    return RadiusGraphPBC(r=radius, loop=loop,
                          max_num_neighbors=max_neighbours)


def get_radius_graph_config(config, loop=False):
    return get_radius_graph(config["radius"], config["max_neighbours"],
                            loop)


def get_radius_graph_pbc_config(config, loop=False):
    return get_radius_graph_pbc(config["radius"],
                                config["max_neighbours"], loop)


def gather_deg(dataset) -> torch.Tensor:
    """In-degree histogram over a dataset (PNA scalers), all-reduced
    across ranks when a process group is up."""
    return _gather_deg(dataset)


gather_deg_dist = gather_deg
gather_deg_mpi = gather_deg


def calculate_avg_deg(dataset) -> float:
    return _calculate_avg_deg(dataset)


def should_skip_self_loops(loop: bool, data: Data) -> bool:
    """Self-loops are skipped unless explicitly requested."""
    return not loop


def check_if_graph_size_variable_dist(*loaders) -> bool:
    return check_if_graph_size_variable(*loaders)


check_if_graph_size_variable_mpi = check_if_graph_size_variable_dist


def check_data_samples_equivalence(data1: Data, data2: Data,
                                   tol: float) -> bool:
    """Edge-order-insensitive sample equality (reference :93): shapes
    match and every edge of data1 appears in data2 with edge_attr
    within tol."""
    if (data1.x.shape != data2.x.shape
            or data1.pos.shape != data2.pos.shape
            or data1.y.shape != data2.y.shape):
        return False
    E = data1.edge_index.shape[1]
    if E != data2.edge_index.shape[1]:
        return False
    # match edges by (src, dst) key
    key1 = (data1.edge_index[0] * (data1.num_nodes + 1)
            + data1.edge_index[1])
    key2 = (data2.edge_index[0] * (data2.num_nodes + 1)
            + data2.edge_index[1])
    order1, order2 = torch.argsort(key1), torch.argsort(key2)
    if not torch.equal(key1[order1], key2[order2]):
        return False
    a1 = getattr(data1, "edge_attr", None)
    a2 = getattr(data2, "edge_attr", None)
    if a1 is not None and a2 is not None:
        if (a1[order1] - a2[order2]).norm(dim=-1).max() >= tol:
            return False
    return True


def update_predicted_values(type: List[str], index: List[int],
                            graph_feature_dim: List[int],
                            node_feature_dim: List[int], data: Data):
    """Assemble data.y and data.y_loc from selected graph/node features
    (reference :604): the concatenated output vector the multi-head
    loss slices by head."""
    output_feature = []
    data.y_loc = torch.zeros(1, len(type) + 1, dtype=torch.int64,
                             device=data.x.device)
    raw_y = data.y
    for item in range(len(type)):
        if type[item] == "graph":
            start = sum(graph_feature_dim[:index[item]])
            feat_ = raw_y.flatten()[
                start:start + graph_feature_dim[index[item]]
            ].reshape(graph_feature_dim[index[item]], 1)
        elif type[item] == "node":
            start = sum(node_feature_dim[:index[item]])
            feat_ = data.x[
                :, start:start + node_feature_dim[index[item]]
            ].reshape(-1, 1)
        else:
            raise ValueError("Unknown output type", type[item])
        output_feature.append(feat_)
        data.y_loc[0, item + 1] = (data.y_loc[0, item]
                                   + feat_.shape[0] * feat_.shape[1])
    data.y = torch.cat(output_feature, dim=0)
    return data


def update_atom_features(atom_features, data: Data):
    """Restrict data.x to the selected feature columns (reference
    :648; accepts AtomFeatures enum members or ints)."""
    idx = [f.value if hasattr(f, "value") else int(f)
           for f in atom_features]
    data.x = data.x[:, idx]
    return data


# reference-named aliases (reference splits dist/MPI variants;
# ours picks the plane via HYDRAGNN_AGGR_BACKEND)
gather_deg_dist = gather_deg
gather_deg_mpi = gather_deg
check_if_graph_size_variable_mpi = check_if_graph_size_variable_dist
