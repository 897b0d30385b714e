"""Lennard-Jones MLIP example with PBC (reference
examples/LennardJones/LennardJones.py:56-345 + LJ_data.py:53-450):
synthetic periodic LJ configurations with analytic energies/forces,
trained with the interatomic-potential wrapper."""

import argparse
import json
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from hydragnn_amd.preprocess import create_dataloaders, split_dataset
from hydragnn_amd.models import create_model_config
from hydragnn_amd.train import train_validate_test
from hydragnn_amd.utils.config import get_log_name_config, update_config
from hydragnn_amd.utils.datasets.synthetic import lj_dataset
from hydragnn_amd.utils.distributed import setup_ddp, distributed_model_wrapper
from hydragnn_amd.utils.model import save_model
from hydragnn_amd.utils.optimizer import select_optimizer


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--mpnn_type", default=None)
    parser.add_argument("--num_epoch", type=int, default=None)
    parser.add_argument("--num_samples", type=int, default=64)
    args = parser.parse_args()

    config_path = os.path.join(os.path.dirname(__file__),
                               "LennardJones.json")
    with open(config_path) as f:
        config = json.load(f)
    if args.mpnn_type:
        config["NeuralNetwork"]["Architecture"]["mpnn_type"] = args.mpnn_type
    if args.num_epoch:
        config["NeuralNetwork"]["Training"]["num_epoch"] = args.num_epoch

    setup_ddp()
    torch.manual_seed(11)
    arch = config["NeuralNetwork"]["Architecture"]
    dataset = lj_dataset(num_samples=args.num_samples, num_atoms=32,
                         radius=arch["radius"], pbc=True)
    trainset, valset, testset = split_dataset(
        dataset, config["NeuralNetwork"]["Training"]["perc_train"])
    loaders = create_dataloaders(
        trainset, valset, testset,
        config["NeuralNetwork"]["Training"]["batch_size"], config=config)
    config = update_config(config, *loaders)

    log_name = get_log_name_config(config)
    model = create_model_config(config["NeuralNetwork"])
    model = distributed_model_wrapper(model)
    optimizer = select_optimizer(
        model, config["NeuralNetwork"]["Training"]["Optimizer"])
    train_validate_test(model, optimizer, *loaders, writer=None,
                        scheduler=None, config=config["NeuralNetwork"],
                        log_name=log_name,
                        verbosity=config["Verbosity"]["level"])
    save_model(model, optimizer, log_name)


if __name__ == "__main__":
    main()
