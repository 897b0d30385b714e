"""Per-sample pickle datasets (reference: hydragnn/utils/datasets/
pickledataset.py:24-113 SimplePickleDataset/Writer)."""

from __future__ import annotations

import os
import pickle

import torch.distributed as dist

from .abstractbasedataset import AbstractBaseDataset


class SimplePickleWriter:
    """Write each sample as its own pickle file + a meta pickle
    (minmax, attrs)."""

    def __init__(self, dataset, basedir: str, label: str = "total",
                 minmax_node_feature=None, minmax_graph_feature=None,
                 use_subdir: bool = False, attrs: dict | None = None):
        rank = dist.get_rank() if dist.is_initialized() else 0
        nranks = dist.get_world_size() if dist.is_initialized() else 1
        os.makedirs(basedir, exist_ok=True)
        if rank == 0:
            meta = {
                "minmax_node_feature": minmax_node_feature,
                "minmax_graph_feature": minmax_graph_feature,
                "ndata": len(dataset) * nranks,
                "use_subdir": use_subdir,
                "attrs": attrs or {},
            }
            with open(os.path.join(basedir, f"{label}-meta.pkl"), "wb") as f:
                pickle.dump(meta, f)
        if dist.is_initialized():
            dist.barrier()
        local_counts = [len(dataset)] * 1
        start = rank * len(dataset)
        for i, data in enumerate(dataset):
            gid = start + i
            subdir = ""
            if use_subdir:
                subdir = str(gid // 1000)
                os.makedirs(os.path.join(basedir, subdir), exist_ok=True)
            fname = os.path.join(basedir, subdir, f"{label}-{gid}.pkl")
            with open(fname, "wb") as f:
                pickle.dump(data, f)
        if dist.is_initialized():
            dist.barrier()


class SimplePickleDataset(AbstractBaseDataset):
    def __init__(self, basedir: str, label: str = "total", subset=None):
        super().__init__()
        self.basedir = basedir
        self.label = label
        with open(os.path.join(basedir, f"{label}-meta.pkl"), "rb") as f:
            meta = pickle.load(f)
        self.ndata = meta["ndata"]
        self.use_subdir = meta.get("use_subdir", False)
        self.minmax_node_feature = meta.get("minmax_node_feature")
        self.minmax_graph_feature = meta.get("minmax_graph_feature")
        for k, v in meta.get("attrs", {}).items():
            setattr(self, k, v)
        self.subset = subset if subset is not None else list(range(self.ndata))

    def len(self):
        return len(self.subset)

    def get(self, idx):
        gid = self.subset[idx]
        subdir = str(gid // 1000) if self.use_subdir else ""
        fname = os.path.join(self.basedir, subdir,
                             f"{self.label}-{gid}.pkl")
        with open(fname, "rb") as f:
            return pickle.load(f)

    def setsubset(self, subset):
        self.subset = list(subset)
