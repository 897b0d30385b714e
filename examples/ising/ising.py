"""Ising-model example (reference examples/ising): lattice spin
configurations with a closed-form Hamiltonian as the graph target."""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from hydragnn_amd.data import Data
from hydragnn_amd.models import create_model_config
from hydragnn_amd.ops import radius_graph, scatter
from hydragnn_amd.preprocess import create_dataloaders, split_dataset
from hydragnn_amd.train import train_validate_test
from hydragnn_amd.utils.config import update_config
from hydragnn_amd.utils.distributed import setup_ddp, distributed_model_wrapper
from hydragnn_amd.utils.optimizer import select_optimizer


def ising_dataset(num_samples=128, lattice=4, J=1.0, seed=17):
    """Spins on a cubic lattice; E = -J sum_<ij> s_i s_j."""
    g = torch.Generator().manual_seed(seed)
    grid = torch.stack(torch.meshgrid(
        torch.arange(lattice), torch.arange(lattice),
        torch.arange(lattice), indexing="ij"), -1).reshape(-1, 3).float()
    n = grid.shape[0]
    ei = radius_graph(grid, 1.1, max_num_neighbors=6)
    ds = []
    for _ in range(num_samples):
        s = torch.randint(0, 2, (n, 1), generator=g).float() * 2 - 1
        e_pair = s[ei[0]] * s[ei[1]]
        energy = -J * 0.5 * e_pair.sum() / n  # per-site energy
        d = Data(x=s, pos=grid.clone(), edge_index=ei,
                 y=energy.view(1, 1),
                 y_loc=torch.tensor([[0, 1]]))
        d.num_nodes = n
        ds.append(d)
    return ds


CONFIG = {
    "Verbosity": {"level": 0},
    "Dataset": {"name": "ising_synthetic"},
    "NeuralNetwork": {
        "Architecture": {
            "mpnn_type": "GIN", "radius": 1.1, "max_neighbours": 6,
            "hidden_dim": 32, "num_conv_layers": 2,
            "output_heads": {"graph": {
                "num_sharedlayers": 1, "dim_sharedlayers": 32,
                "num_headlayers": 2, "dim_headlayers": [32, 32]}},
            "task_weights": [1.0],
        },
        "Variables_of_interest": {
            "input_node_features": [0],
            "output_names": ["energy_per_site"], "output_index": [0],
            "type": ["graph"], "denormalize_output": False,
        },
        "Training": {
            "num_epoch": 15, "perc_train": 0.8, "batch_size": 32,
            "loss_function_type": "mse", "EarlyStopping": False,
            "Checkpoint": False,
            "Optimizer": {"type": "AdamW", "learning_rate": 0.005},
        },
    },
}


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--num_epoch", type=int, default=None)
    args = parser.parse_args()
    config = dict(CONFIG)
    if args.num_epoch:
        config["NeuralNetwork"]["Training"]["num_epoch"] = args.num_epoch
    setup_ddp()
    torch.manual_seed(17)
    dataset = ising_dataset()
    splits = split_dataset(
        dataset, config["NeuralNetwork"]["Training"]["perc_train"])
    loaders = create_dataloaders(*splits, 32, config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"])
    model = distributed_model_wrapper(model)
    opt = select_optimizer(
        model, config["NeuralNetwork"]["Training"]["Optimizer"])
    train_validate_test(model, opt, *loaders, writer=None, scheduler=None,
                        config=config["NeuralNetwork"],
                        log_name="ising", verbosity=0)


if __name__ == "__main__":
    main()
