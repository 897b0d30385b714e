"""Raw-dataset processing pipeline.

Behavioral parity with /root/reference/hydragnn/utils/datasets/
abstractrawdataset.py:39-415: distributed file-list sharding (nsplit
per rank), per-file transform_input_to_data_object_base, feature
scaling by num-nodes, min-max normalization with distributed MIN/MAX
reduction, optional rotation normalization, radius-graph build (+PBC),
edge-length attributes normalized by the globally-reduced max edge
length, stratified subsampling.
"""

from __future__ import annotations

import os
from abc import abstractmethod
from typing import List, Optional

import torch
import torch.distributed as dist

from ...data import Data
from ...ops import get_edge_vectors_and_lengths, radius_graph, radius_graph_pbc
from ...preprocess.transforms import add_edge_lengths, normalize_rotation
from ..distributed import nsplit
from .abstractbasedataset import AbstractBaseDataset


class AbstractRawDataset(AbstractBaseDataset):
    def __init__(self, config, dist_flag: bool = False, sampling=None):
        super().__init__()
        self.config = config
        self.dist = dist_flag
        self.sampling = sampling
        ds = config["Dataset"]
        self.raw_dataset_name = ds["name"]
        self.data_format = ds.get("format", "LSMS")
        self.path_dictionary = ds["path"]
        nf = ds.get("node_features", {})
        gf = ds.get("graph_features", {})
        self.node_feature_name = nf.get("name", [])
        self.node_feature_dim = nf.get("dim", [])
        self.node_feature_col = nf.get("column_index", [])
        self.graph_feature_name = gf.get("name", [])
        self.graph_feature_dim = gf.get("dim", [])
        self.graph_feature_col = gf.get("column_index", [])
        arch = config["NeuralNetwork"]["Architecture"]
        self.radius = arch.get("radius", 5.0)
        self.max_neighbours = arch.get("max_neighbours", 100)
        self.periodic = arch.get("periodic_boundary_conditions", False)
        self.rotational_invariance = ds.get("rotational_invariance", False)
        self.minmax_node_feature = None
        self.minmax_graph_feature = None
        self._load_raw_data()

    # -- abstract ----------------------------------------------------------
    @abstractmethod
    def transform_input_to_data_object_base(self, filepath) -> Optional[Data]:
        ...

    # -- pipeline ----------------------------------------------------------
    def _rank_info(self):
        if self.dist and dist.is_initialized():
            return dist.get_world_size(), dist.get_rank()
        return 1, 0

    def _load_raw_data(self):
        for dataset_type, raw_path in self.path_dictionary.items():
            if not os.path.isdir(raw_path):
                raise ValueError(f"Raw dataset path not found: {raw_path}")
            files = sorted(os.listdir(raw_path))
            world, rank = self._rank_info()
            if world > 1:
                files = list(nsplit(files, world))[rank]
            for name in files:
                filepath = os.path.join(raw_path, name)
                data = self.transform_input_to_data_object_base(filepath)
                if data is not None:
                    self.dataset.append(data)

        self._scale_features_by_num_nodes()
        self.normalize_dataset()
        if self.rotational_invariance:
            for d in self.dataset:
                normalize_rotation(d)
        self._build_edges()
        if self.sampling is not None and self.sampling < 1.0:
            from ...preprocess.stratified_sampling import stratified_sampling
            self.dataset = stratified_sampling(self.dataset, self.sampling)

    def _scale_features_by_num_nodes(self):
        """Divide extensive features by num_nodes
        (reference :305-328, names ending in _scaled_num_nodes)."""
        scaled_graph = [i for i, name in enumerate(self.graph_feature_name)
                        if name and name.endswith("_scaled_num_nodes")]
        scaled_node = [i for i, name in enumerate(self.node_feature_name)
                       if name and name.endswith("_scaled_num_nodes")]
        for d in self.dataset:
            n = d.num_nodes
            for i in scaled_graph:
                d.y[i] = d.y[i] / n
            for i in scaled_node:
                c0 = sum(self.node_feature_dim[:i])
                c1 = c0 + self.node_feature_dim[i]
                d.x[:, c0:c1] = d.x[:, c0:c1] / n

    def normalize_dataset(self):
        """Min-max normalize node/graph features with distributed
        MIN/MAX reduction (reference :217-299)."""
        if not self.dataset:
            return
        num_node_f = self.dataset[0].x.shape[1] if \
            self.dataset[0].get("x") is not None else 0
        num_graph_f = self.dataset[0].y.numel() if \
            self.dataset[0].get("y") is not None else 0
        nmin = torch.full((num_node_f,), float("inf"))
        nmax = torch.full((num_node_f,), float("-inf"))
        gmin = torch.full((num_graph_f,), float("inf"))
        gmax = torch.full((num_graph_f,), float("-inf"))
        for d in self.dataset:
            if num_node_f:
                nmin = torch.minimum(nmin, d.x.min(dim=0).values)
                nmax = torch.maximum(nmax, d.x.max(dim=0).values)
            if num_graph_f:
                gmin = torch.minimum(gmin, d.y.view(-1))
                gmax = torch.maximum(gmax, d.y.view(-1))
        if self.dist and dist.is_initialized():
            from ..distributed import to_comm_device
            for t, op in ((nmin, dist.ReduceOp.MIN),
                          (nmax, dist.ReduceOp.MAX),
                          (gmin, dist.ReduceOp.MIN),
                          (gmax, dist.ReduceOp.MAX)):
                if t.numel():
                    tt, moved = to_comm_device(t)
                    dist.all_reduce(tt, op=op)
                    if moved:
                        t.copy_(tt.cpu())
        for d in self.dataset:
            if num_node_f:
                span = (nmax - nmin).clamp(min=1e-12)
                d.x = (d.x - nmin) / span
            if num_graph_f:
                span = (gmax - gmin).clamp(min=1e-12)
                d.y = ((d.y.view(-1) - gmin) / span).view_as(d.y)
        self.minmax_node_feature = torch.stack([nmin, nmax]).numpy() \
            if num_node_f else None
        self.minmax_graph_feature = torch.stack([gmin, gmax]).numpy() \
            if num_graph_f else None

    def _build_edges(self):
        max_len = torch.zeros(1)
        for d in self.dataset:
            if self.periodic and d.get("supercell_size") is not None:
                ei, shifts = radius_graph_pbc(
                    d.pos, self.radius, d.supercell_size,
                    max_num_neighbors=self.max_neighbours)
                d.edge_index = ei
                d.edge_shifts = shifts
            else:
                d.edge_index = radius_graph(
                    d.pos, self.radius,
                    max_num_neighbors=self.max_neighbours)
            if d.num_edges > 0:
                _, lengths = get_edge_vectors_and_lengths(
                    d.pos, d.edge_index, d.get("edge_shifts"))
                max_len = torch.maximum(max_len, lengths.max().view(1))
        if self.dist and dist.is_initialized():
            from ..distributed import to_comm_device
            max_len, _ = to_comm_device(max_len)
            dist.all_reduce(max_len, op=dist.ReduceOp.MAX)
            max_len = max_len.cpu()
        scale = float(max_len.clamp(min=1e-12))
        for d in self.dataset:
            add_edge_lengths(d, max_length=scale)

    # -- Dataset API -------------------------------------------------------
    def len(self):
        return len(self.dataset)

    def get(self, idx):
        return self.dataset[idx]
