"""Serialized whole-dataset pickle (reference:
hydragnn/utils/datasets/serializeddataset.py:20-97)."""

from __future__ import annotations

import os
import pickle

from .abstractbasedataset import AbstractBaseDataset


class SerializedWriter:
    def __init__(self, dataset, basedir: str, label: str = "total",
                 minmax_node_feature=None, minmax_graph_feature=None):
        os.makedirs(basedir, exist_ok=True)
        with open(os.path.join(basedir, f"{label}.pkl"), "wb") as f:
            pickle.dump(minmax_node_feature, f)
            pickle.dump(minmax_graph_feature, f)
            pickle.dump(list(dataset), f)


class SerializedDataset(AbstractBaseDataset):
    def __init__(self, basedir: str, label: str = "total"):
        super().__init__()
        with open(os.path.join(basedir, f"{label}.pkl"), "rb") as f:
            self.minmax_node_feature = pickle.load(f)
            self.minmax_graph_feature = pickle.load(f)
            self.dataset = pickle.load(f)

    def len(self):
        return len(self.dataset)

    def get(self, idx):
        return self.dataset[idx]
