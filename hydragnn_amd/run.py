"""Convenience entry points: run_training / run_prediction.

The reference removed these in v5 in favor of explicit orchestration
(README.md:144-147 there), but the JSON-config one-call API remains the
historical front door — provided here as thin wrappers over the same
explicit flow (create_dataloaders -> update_config ->
create_model_config -> distributed_model_wrapper ->
train_validate_test).
"""

from __future__ import annotations

import json
import os
from typing import Union

import torch

from .models import create_model_config
from .preprocess import create_dataloaders, split_dataset
from .preprocess.load_data import dataset_loading_and_splitting
from .train import train_validate_test, test as test_fn
from .utils.config import get_log_name_config, save_config, update_config
from .utils.distributed import distributed_model_wrapper, setup_ddp
from .utils.model import (get_summary_writer, load_existing_model,
                          load_existing_model_config, save_model)
from .utils.optimizer import select_optimizer
from .utils.print.print_utils import setup_log


def _load_config(config: Union[str, dict]) -> dict:
    if isinstance(config, str):
        with open(config) as f:
            return json.load(f)
    return config


def run_training(config: Union[str, dict], dataset=None,
                 use_gpu: bool = True):
    """Train from a JSON config (path or dict).  `dataset` may be a
    list of Data samples (split internally) or None to use the
    config-driven dataset loading."""
    config = _load_config(config)
    setup_ddp()
    verbosity = config.get("Verbosity", {}).get("level", 0)

    if dataset is not None:
        splits = split_dataset(
            dataset, config["NeuralNetwork"]["Training"]["perc_train"],
            stratify_splitting=config.get("Dataset", {}).get(
                "compositional_stratified_splitting", False))
        loaders = create_dataloaders(
            *splits, config["NeuralNetwork"]["Training"]["batch_size"],
            config=config)
    else:
        loaders = dataset_loading_and_splitting(config)

    config = update_config(config, *loaders)
    log_name = get_log_name_config(config)
    setup_log(log_name)
    save_config(config, log_name)

    model = create_model_config(config["NeuralNetwork"],
                                verbosity=verbosity, use_gpu=use_gpu)
    optimizer = select_optimizer(
        model, config["NeuralNetwork"]["Training"]["Optimizer"])
    # restart support (reference model.py:204-211)
    load_existing_model_config(
        model, config["NeuralNetwork"]["Training"], optimizer=optimizer)
    model = distributed_model_wrapper(
        model, verbosity=verbosity,
        sync_batch_norm=config["NeuralNetwork"]["Architecture"].get(
            "SyncBatchNorm", False))
    scheduler = torch.optim.lr_scheduler.ReduceLROnPlateau(
        optimizer, mode="min", factor=0.5, patience=5)
    writer = get_summary_writer(log_name)

    train_validate_test(model, optimizer, *loaders, writer=writer,
                        scheduler=scheduler,
                        config=config["NeuralNetwork"],
                        log_name=log_name, verbosity=verbosity,
                        create_plots=config.get("Visualization", {})
                        .get("create_plots", False))
    save_model(model, optimizer, log_name)
    return model, config


def run_prediction(config: Union[str, dict], model=None, dataset=None,
                   use_gpu: bool = True):
    """Evaluate a trained model from a JSON config; loads the
    checkpoint named by the config when `model` is None.  Returns
    (error, per-task errors, true values, predicted values)."""
    config = _load_config(config)
    setup_ddp()
    verbosity = config.get("Verbosity", {}).get("level", 0)
    if dataset is not None:
        splits = split_dataset(
            dataset, config["NeuralNetwork"]["Training"]["perc_train"])
        loaders = create_dataloaders(
            *splits, config["NeuralNetwork"]["Training"]["batch_size"],
            config=config)
    else:
        loaders = dataset_loading_and_splitting(config)
    config = update_config(config, *loaders)
    if model is None:
        model = create_model_config(config["NeuralNetwork"],
                                    verbosity=verbosity, use_gpu=use_gpu)
        log_name = get_log_name_config(config)
        load_existing_model(model, log_name)
    return test_fn(loaders[2], model, verbosity)
