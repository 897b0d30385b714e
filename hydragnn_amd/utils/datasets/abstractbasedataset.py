"""Dataset base classes (reference: hydragnn/utils/datasets/
abstractbasedataset.py:16-77)."""

from __future__ import annotations

from abc import ABC, abstractmethod

import torch

# name -> stable integer ID used for per-dataset branch masking
# (reference abstractbasedataset.py:52-77)
DATASET_NAME_TO_ID = {}


def dataset_name_to_id(name: str) -> int:
    if name not in DATASET_NAME_TO_ID:
        DATASET_NAME_TO_ID[name] = len(DATASET_NAME_TO_ID)
    return DATASET_NAME_TO_ID[name]


class AbstractBaseDataset(torch.utils.data.Dataset, ABC):
    """List-backed dataset of hydragnn_amd.data.Data samples."""

    def __init__(self):
        super().__init__()
        self.dataset = []

    @abstractmethod
    def get(self, idx):
        ...

    @abstractmethod
    def len(self):
        ...

    def __len__(self):
        return self.len()

    def __getitem__(self, idx):
        return self.get(idx)

    def apply(self, fn):
        for d in self.dataset:
            fn(d)
