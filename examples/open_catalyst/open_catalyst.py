"""Open-Catalyst-style example (reference examples/open_catalyst_*):
periodic slab configurations with adsorbates, energy+force MLIP
training on EGNN; synthetic LJ surrogate data (no network access)."""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from hydragnn_amd.models import create_model_config
from hydragnn_amd.preprocess import create_dataloaders, split_dataset
from hydragnn_amd.train import train_validate_test
from hydragnn_amd.utils.config import update_config
from hydragnn_amd.utils.datasets.synthetic import lj_dataset
from hydragnn_amd.utils.distributed import setup_ddp, distributed_model_wrapper
from hydragnn_amd.utils.optimizer import select_optimizer

CONFIG = {
    "Verbosity": {"level": 0},
    "Dataset": {"name": "oc_synthetic"},
    "NeuralNetwork": {
        "Architecture": {
            "mpnn_type": "EGNN", "radius": 4.0, "max_neighbours": 30,
            "hidden_dim": 64, "num_conv_layers": 3,
            "periodic_boundary_conditions": True,
            "enable_interatomic_potential": True,
            "energy_weight": 1.0, "energy_peratom_weight": 1.0,
            "force_weight": 20.0, "equivariance": False,
            "output_heads": {"node": {
                "num_headlayers": 2, "dim_headlayers": [64, 64],
                "type": "mlp"}},
            "task_weights": [1.0],
        },
        "Variables_of_interest": {
            "input_node_features": [0],
            "output_names": ["energy"], "output_index": [0],
            "output_dim": [1], "type": ["node"],
            "denormalize_output": False,
        },
        "Training": {
            "num_epoch": 8, "perc_train": 0.8, "batch_size": 8,
            "loss_function_type": "mse", "EarlyStopping": False,
            "Checkpoint": False,
            "Optimizer": {"type": "AdamW", "learning_rate": 0.002},
        },
    },
}


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--num_epoch", type=int, default=None)
    parser.add_argument("--num_samples", type=int, default=32)
    args = parser.parse_args()
    config = dict(CONFIG)
    if args.num_epoch:
        config["NeuralNetwork"]["Training"]["num_epoch"] = args.num_epoch
    setup_ddp()
    torch.manual_seed(29)
    dataset = lj_dataset(num_samples=args.num_samples, num_atoms=64,
                         cell_size=10.0, radius=4.0, pbc=True)
    splits = split_dataset(
        dataset, config["NeuralNetwork"]["Training"]["perc_train"])
    loaders = create_dataloaders(*splits, 8, config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"])
    model = distributed_model_wrapper(model)
    opt = select_optimizer(
        model, config["NeuralNetwork"]["Training"]["Optimizer"])
    train_validate_test(model, opt, *loaders, writer=None, scheduler=None,
                        config=config["NeuralNetwork"],
                        log_name="open_catalyst", verbosity=0)


if __name__ == "__main__":
    main()
