"""Shared end-to-end training harness for tests (pattern:
reference tests/_training_workflow.py:57-191)."""

from __future__ import annotations

import torch

from hydragnn_amd.models import create_model_config
from hydragnn_amd.preprocess import create_dataloaders, split_dataset
from hydragnn_amd.train import train_validate_test
from hydragnn_amd.utils.config import update_config
from hydragnn_amd.utils.optimizer import select_optimizer

from deterministic_graph_data import base_config, make_deterministic_dataset


def run_training(mpnn_type, heads=("graph",), num_samples=64,
                 num_epoch=30, overrides=None, dataset_kwargs=None,
                 use_gpu=False, train_kwargs=None):
    torch.manual_seed(7)
    config = base_config(mpnn_type, heads=heads, num_epoch=num_epoch)
    if overrides:
        from hydragnn_amd.utils.config import merge_config
        config = merge_config(config, overrides)
    ds_kwargs = dict(num_heads_node=1 if "node" in heads else 0,
                     include_graph_head="graph" in heads)
    if dataset_kwargs:
        ds_kwargs.update(dataset_kwargs)
    dataset = make_deterministic_dataset(num_samples=num_samples, **ds_kwargs)
    trainset, valset, testset = split_dataset(
        dataset, config["NeuralNetwork"]["Training"]["perc_train"], seed=0)
    train_loader, val_loader, test_loader = create_dataloaders(
        trainset, valset, testset,
        config["NeuralNetwork"]["Training"]["batch_size"], config=config)
    config = update_config(config, train_loader, val_loader, test_loader)
    model = create_model_config(config["NeuralNetwork"], use_gpu=use_gpu)
    optimizer = select_optimizer(
        model, config["NeuralNetwork"]["Training"]["Optimizer"])
    train_validate_test(
        model, optimizer, train_loader, val_loader, test_loader,
        writer=None, scheduler=None, config=config["NeuralNetwork"],
        log_name=f"test_{mpnn_type}", verbosity=0,
        **(train_kwargs or {}))
    return model, config, (train_loader, val_loader, test_loader)


def evaluate_error(model, loader, config):
    """RMSE per head on a loader."""
    from hydragnn_amd.train import test as test_fn
    err, tasks_err, tv, pv = test_fn(loader, model, 0)
    rmses = []
    for t, p in zip(tv, pv):
        if t.numel() == 0:
            rmses.append(float("nan"))
        else:
            rmses.append(float(torch.sqrt(((t - p) ** 2).mean())))
    return float(err), rmses
