"""LSMS raw-format example (reference examples/lsms): writes synthetic
LSMS text files, then runs the FULL raw pipeline — LSMSDataset parsing,
min-max normalization, radius-graph build with normalized edge-length
attributes — into training."""

import argparse
import os
import sys
import tempfile

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from hydragnn_amd.models import create_model_config
from hydragnn_amd.preprocess import create_dataloaders, split_dataset
from hydragnn_amd.train import train_validate_test
from hydragnn_amd.utils.config import update_config
from hydragnn_amd.utils.datasets.rawloaders import LSMSDataset
from hydragnn_amd.utils.distributed import setup_ddp, distributed_model_wrapper
from hydragnn_amd.utils.optimizer import select_optimizer


def write_raw_lsms(raw_dir, num_samples=64, n_atoms=16, seed=23):
    """Synthetic FePt-style LSMS files: header = [free_energy, _],
    atom rows = idx 0 x y z Z charge."""
    g = torch.Generator().manual_seed(seed)
    os.makedirs(raw_dir, exist_ok=True)
    for isample in range(num_samples):
        pos = torch.rand(n_atoms, 3, generator=g) * 4.0
        z = torch.where(torch.rand(n_atoms, generator=g) < 0.5,
                        torch.tensor(26.0), torch.tensor(78.0))
        charge = z + torch.randn(n_atoms, generator=g) * 0.05
        energy = float((z == 26).float().mean() * 2.0 - 1.0
                       + 0.1 * torch.randn(1, generator=g))
        lines = [f"{energy:.6f} 0.0"]
        for i in range(n_atoms):
            lines.append(
                f"{i} 0 {pos[i, 0]:.6f} {pos[i, 1]:.6f} {pos[i, 2]:.6f} "
                f"{z[i]:.1f} {charge[i]:.6f}")
        with open(os.path.join(raw_dir, f"cfg_{isample:04d}.txt"),
                  "w") as f:
            f.write("\n".join(lines) + "\n")


def build_config(raw_dir):
    return {
        "Verbosity": {"level": 0},
        "Dataset": {
            "name": "FePt_synthetic",
            "format": "LSMS",
            "path": {"total": raw_dir},
            "node_features": {"name": ["num_of_protons",
                                       "charge_density"],
                              "dim": [1, 1], "column_index": [5, 6]},
            "graph_features": {"name": ["free_energy"], "dim": [1],
                               "column_index": [0]},
        },
        "NeuralNetwork": {
            "Architecture": {
                "mpnn_type": "CGCNN", "radius": 2.0,
                "max_neighbours": 12, "hidden_dim": 2,
                "num_conv_layers": 2, "edge_features": ["lengths"],
                "output_heads": {"graph": {
                    "num_sharedlayers": 1, "dim_sharedlayers": 16,
                    "num_headlayers": 2, "dim_headlayers": [16, 16]}},
                "task_weights": [1.0],
            },
            "Variables_of_interest": {
                "input_node_features": [0, 1],
                "output_names": ["free_energy"], "output_index": [0],
                "output_dim": [1], "type": ["graph"],
                "denormalize_output": False,
            },
            "Training": {
                "num_epoch": 10, "perc_train": 0.8, "batch_size": 16,
                "loss_function_type": "mse", "EarlyStopping": False,
                "Checkpoint": False,
                "Optimizer": {"type": "AdamW", "learning_rate": 0.01},
            },
        },
    }


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--num_epoch", type=int, default=None)
    parser.add_argument("--raw_dir", default=None)
    args = parser.parse_args()
    raw_dir = args.raw_dir or os.path.join(tempfile.mkdtemp(), "raw")
    write_raw_lsms(raw_dir)
    config = build_config(raw_dir)
    if args.num_epoch:
        config["NeuralNetwork"]["Training"]["num_epoch"] = args.num_epoch
    setup_ddp()
    torch.manual_seed(23)
    dataset = LSMSDataset(config)
    for d in dataset:
        d.y = d.y.view(1, 1)
        d.y_loc = torch.tensor([[0, 1]])
    splits = split_dataset(
        dataset, config["NeuralNetwork"]["Training"]["perc_train"])
    loaders = create_dataloaders(*splits, 16, config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"])
    model = distributed_model_wrapper(model)
    opt = select_optimizer(
        model, config["NeuralNetwork"]["Training"]["Optimizer"])
    train_validate_test(model, opt, *loaders, writer=None, scheduler=None,
                        config=config["NeuralNetwork"],
                        log_name="lsms", verbosity=0)


if __name__ == "__main__":
    main()
