"""Dataset splitting and dataloader creation.

Reference: hydragnn/preprocess/load_data.py:234-516 —
create_dataloaders with DistributedSampler (or distributed cost-aware
node-budget batch sampler), split_dataset with stratified option.
The loader collates our Data into Batch (data.py) and sorts edges by
destination so the HIP segment kernels get CSR rows for free.
"""

from __future__ import annotations

import os
import random
from typing import List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist
from torch.utils.data import DataLoader as TorchDataLoader
from torch.utils.data.distributed import DistributedSampler

from ..data import Batch, Data
from .batch_sampler import CostAwareBatchSampler, DistributedCostAwareBatchSampler


def _collate(data_list: Sequence[Data]) -> Batch:
    return Batch.from_data_list(list(data_list))


def split_dataset(dataset, perc_train: float,
                  stratify_splitting: bool = False, seed: int = 0):
    """Random (or composition-stratified) train/val/test split:
    train = perc_train, val = test = (1-perc_train)/2."""
    n = len(dataset)
    idx = list(range(n))
    rng = random.Random(seed)
    if stratify_splitting:
        from .compositional_splitting import compositional_stratified_split
        return compositional_stratified_split(dataset, perc_train, seed)
    rng.shuffle(idx)
    n_train = int(n * perc_train)
    n_val = int(n * (1.0 - perc_train) / 2.0)
    train_idx = idx[:n_train]
    val_idx = idx[n_train:n_train + n_val]
    test_idx = idx[n_train + n_val:]
    take = lambda ids: [dataset[i] for i in ids]
    return take(train_idx), take(val_idx), take(test_idx)


def create_dataloaders(trainset, valset, testset, batch_size: int,
                       sampler_shuffle: bool = True, group=None,
                       oversampling: bool = False,
                       num_samples: Optional[int] = None,
                       config=None):
    """Build train/val/test loaders.  With torch.distributed
    initialized, uses DistributedSampler; node-budget batching
    (Training.Batching.mode = node_budget) uses the cost-aware
    samplers."""
    num_workers = int(os.getenv("HYDRAGNN_NUM_WORKERS", "0"))
    batching = None
    if config is not None:
        batching = config["NeuralNetwork"]["Training"].get("Batching")

    if oversampling and batching is not None and \
            batching.get("mode") == "node_budget":
        raise ValueError(
            "cost-aware batching cannot be combined with oversampling")

    use_dist = dist.is_initialized() and dist.get_world_size() > 1

    use_custom = os.getenv("HYDRAGNN_CUSTOM_DATALOADER", "0") == "1"

    # Training.Batching.mode == "static_shape": fixed-capacity padded
    # batches (preprocess/static_batch.py) so the train loop's
    # hipGraph-captured step engages — caps computed once over the
    # training set (worst case, shuffle-safe)
    static_coll = None
    if batching is not None and batching.get("mode") == "static_shape":
        from .static_batch import StaticShapeCollater, compute_static_caps
        node_cap, edge_cap = compute_static_caps(
            trainset, batch_size,
            sequential=bool(batching.get("sequential", False)))
        static_coll = StaticShapeCollater(
            node_cap, edge_cap,
            pad_spacing=float(batching.get("pad_spacing", 30.0)))

    def make(ds, shuffle):
        if ds is None or len(ds) == 0:
            return TorchDataLoader([], batch_size=batch_size,
                                   collate_fn=_collate)
        if static_coll is not None and shuffle:
            # padded static batches for the TRAIN loader only (eval
            # loaders keep exact batches)
            if use_dist:
                sampler = DistributedSampler(
                    ds, shuffle=sampler_shuffle)
                return TorchDataLoader(
                    ds, batch_size=batch_size, sampler=sampler,
                    collate_fn=static_coll, num_workers=num_workers,
                    drop_last=True,
                    pin_memory=torch.cuda.is_available())
            return TorchDataLoader(
                ds, batch_size=batch_size, shuffle=sampler_shuffle,
                collate_fn=static_coll, num_workers=num_workers,
                drop_last=True,
                pin_memory=torch.cuda.is_available())
        if use_custom:
            from .dataloader import HydraDataLoader
            bs = None
            if batching is not None and \
                    batching.get("mode") == "node_budget":
                cls = (DistributedCostAwareBatchSampler if use_dist
                       else CostAwareBatchSampler)
                bs = cls(ds, max_nodes=batching["max_nodes"],
                         shuffle=shuffle, seed=batching.get("seed", 0),
                         oversized_policy=batching.get(
                             "oversized_policy", "error"),
                         drop_last=batching.get("drop_last", False))
            return HydraDataLoader(
                ds, batch_size=batch_size, shuffle=shuffle,
                batch_sampler=bs,
                num_workers=max(num_workers, 2))
        if batching is not None and batching.get("mode") == "node_budget":
            max_nodes = batching["max_nodes"]
            if use_dist:
                bs = DistributedCostAwareBatchSampler(
                    ds, max_nodes=max_nodes, shuffle=shuffle,
                    seed=batching.get("seed", 0),
                    oversized_policy=batching.get("oversized_policy",
                                                  "error"),
                    drop_last=batching.get("drop_last", False))
            else:
                bs = CostAwareBatchSampler(
                    ds, max_nodes=max_nodes, shuffle=shuffle,
                    seed=batching.get("seed", 0),
                    oversized_policy=batching.get("oversized_policy",
                                                  "error"),
                    drop_last=batching.get("drop_last", False))
            return TorchDataLoader(ds, batch_sampler=bs,
                                   collate_fn=_collate,
                                   num_workers=num_workers)
        if oversampling and shuffle and not use_dist:
            from torch.utils.data import RandomSampler
            sampler = RandomSampler(
                ds, replacement=True,
                num_samples=num_samples or len(ds))
            return TorchDataLoader(ds, batch_size=batch_size,
                                   sampler=sampler, collate_fn=_collate,
                                   num_workers=num_workers)
        if use_dist:
            sampler = DistributedSampler(ds, shuffle=shuffle and
                                         sampler_shuffle)
            return TorchDataLoader(ds, batch_size=batch_size,
                                   sampler=sampler, collate_fn=_collate,
                                   num_workers=num_workers,
                                   pin_memory=torch.cuda.is_available())
        return TorchDataLoader(ds, batch_size=batch_size,
                               shuffle=shuffle and sampler_shuffle,
                               collate_fn=_collate,
                               num_workers=num_workers,
                               pin_memory=torch.cuda.is_available())

    return make(trainset, True), make(valset, False), make(testset, False)


def dataset_loading_and_splitting(config):
    """Config-driven dataset load + split + loader creation for the
    'unit_test'-style serialized flows (reference load_data.py:214)."""
    ds_config = config["Dataset"]
    fmt = ds_config.get("format", "pickle")
    if fmt == "pickle":
        from ..utils.datasets.pickledataset import SimplePickleDataset
        paths = ds_config["path"]
        trainset = SimplePickleDataset(paths["train"])
        valset = SimplePickleDataset(paths["validate"])
        testset = SimplePickleDataset(paths["test"])
    else:
        raise ValueError(f"Unsupported dataset format {fmt}")
    return create_dataloaders(
        trainset, valset, testset,
        batch_size=config["NeuralNetwork"]["Training"]["batch_size"],
        config=config)


def transform_raw_data_to_serialized(config):
    """Legacy path (reference load_data.py:458): parse a raw dataset by
    format and write the serialized pickle used by later runs."""
    from ..utils.datasets.rawloaders import (CFGDataset, LSMSDataset,
                                             XYZDataset)
    from ..utils.datasets.serializeddataset import SerializedWriter
    fmt = config["Dataset"].get("format", "LSMS").upper()
    cls = {"LSMS": LSMSDataset, "XYZ": XYZDataset,
           "CFG": CFGDataset}.get(fmt)
    if cls is None:
        raise ValueError(f"unknown raw format {fmt}")
    ds = cls(config)
    out = os.path.join(os.environ.get("SERIALIZED_DATA_PATH", "."),
                       "serialized_dataset")
    SerializedWriter(ds, out, config["Dataset"]["name"],
                     minmax_node_feature=ds.minmax_node_feature,
                     minmax_graph_feature=ds.minmax_graph_feature)
    return ds


class SimpleDataLoader(TorchDataLoader):
    """Plain single-process loader over graph batches — the reference's
    naive custom-loader baseline (load_data.py:70); HydraDataLoader is
    the prefetching variant."""

    def __init__(self, dataset, batch_size=1, shuffle=False, **kwargs):
        kwargs.pop("collate_fn", None)
        super().__init__(dataset, batch_size=batch_size,
                         shuffle=shuffle, collate_fn=_collate, **kwargs)


def total_to_train_val_test_pkls(config, isdist=False):
    """Split a 'total' serialized pickle into train/validate/test
    pickles and update config paths (reference load_data.py:475)."""
    import pickle

    from ..utils.datasets.serializeddataset import (SerializedDataset,
                                                    SerializedWriter)
    paths = config["Dataset"]["path"]
    base = os.environ.get("SERIALIZED_DATA_PATH", os.getcwd())
    sdir = f"{base}/serialized_dataset"
    name = config["Dataset"]["name"]
    if str(list(paths.values())[0]).endswith(".pkl"):
        file_dir = str(paths["total"])
        sdir = os.path.dirname(file_dir)
        name = os.path.splitext(os.path.basename(file_dir))[0]
    ds = SerializedDataset(sdir, name)
    perc_train = config["NeuralNetwork"]["Training"]["perc_train"]
    trainset, valset, testset = split_dataset(
        list(ds), perc_train,
        config["Dataset"].get("compositional_stratified_splitting",
                              False))
    out = {}
    for label, subset in (("train", trainset), ("validate", valset),
                          ("test", testset)):
        SerializedWriter(subset, sdir, f"{name}_{label}",
                         minmax_node_feature=getattr(
                             ds, "minmax_node_feature", None),
                         minmax_graph_feature=getattr(
                             ds, "minmax_graph_feature", None))
        out[label] = os.path.join(sdir, f"{name}_{label}.pkl")
    config["Dataset"]["path"] = out
    return config


def load_train_val_test_sets(config, isdist=False):
    """Load the per-split serialized pickles named by the config
    (reference load_data.py:430)."""
    from ..utils.datasets.serializeddataset import SerializedDataset
    base = os.environ.get("SERIALIZED_DATA_PATH", os.getcwd())
    sets = {}
    for split, raw_path in config["Dataset"]["path"].items():
        if str(raw_path).endswith(".pkl"):
            sdir = os.path.dirname(str(raw_path))
            label = os.path.splitext(os.path.basename(str(raw_path)))[0]
        else:
            sdir = f"{base}/serialized_dataset"
            label = f"{config['Dataset']['name']}_{split}"
        sets[split] = SerializedDataset(sdir, label)
    return (sets["train"], sets["validate"], sets["test"])
