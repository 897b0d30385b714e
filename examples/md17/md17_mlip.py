"""MD17 MLIP training example — the north-star config (reference
examples/md17/md17_mlip.py:31-120): pre-transform sets x = Z,
y = energy/len(x), forces; radius graph (r=7, max 30) built per sample;
enable_interatomic_potential with energy + energy/atom + force loss.

Synthetic MD17-shaped (aspirin, 21 atoms) data with analytic LJ-form
energies/forces stands in for the real trajectory (no network access).
"""

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
from hydragnn_amd.utils.datasets.synthetic import md17_shape_dataset
from hydragnn_amd.utils.distributed import setup_ddp, distributed_model_wrapper
from hydragnn_amd.utils.model import save_model
from hydragnn_amd.utils.optimizer import select_optimizer


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--mpnn_type", default=None)
    parser.add_argument("--num_epoch", type=int, default=None)
    parser.add_argument("--num_samples", type=int, default=128)
    args = parser.parse_args()

    with open(os.path.join(os.path.dirname(__file__),
                           "md17_mlip.json")) as f:
        config = json.load(f)
    if args.mpnn_type:
        config["NeuralNetwork"]["Architecture"]["mpnn_type"] = args.mpnn_type
    if args.num_epoch:
        config["NeuralNetwork"]["Training"]["num_epoch"] = args.num_epoch

    setup_ddp()
    torch.manual_seed(13)
    arch = config["NeuralNetwork"]["Architecture"]
    dataset = md17_shape_dataset(
        num_samples=args.num_samples, radius=arch["radius"],
        max_neighbours=arch["max_neighbours"])
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
