"""QM9-style multi-head training example — the canonical flow
(reference examples/qm9/qm9.py:60-168): parse JSON config -> build
dataset -> create_dataloaders -> update_config -> create_model_config ->
distributed_model_wrapper -> train_validate_test -> save_model.

This image has no network access, so the dataset is synthetic
QM9-shaped molecules (random small organics with closed-form graph +
node targets); swap in a real QM9 loader by replacing build_dataset().
"""

import argparse
import json
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import hydragnn_amd  # noqa: E402
from hydragnn_amd.preprocess import create_dataloaders, split_dataset
from hydragnn_amd.models import create_model_config
from hydragnn_amd.train import train_validate_test
from hydragnn_amd.utils.config import get_log_name_config, save_config, update_config
from hydragnn_amd.utils.distributed import setup_ddp, distributed_model_wrapper
from hydragnn_amd.utils.model import get_summary_writer, save_model
from hydragnn_amd.utils.optimizer import select_optimizer
from hydragnn_amd.utils.print.print_utils import setup_log


def build_dataset(num_samples=200, seed=5):
    """QM9-shaped synthetic molecules: <=9 heavy atoms + H, graph
    target = synthetic 'internal energy'-like closed form, node target
    = per-atom contribution."""
    from hydragnn_amd.data import Data
    from hydragnn_amd.ops import radius_graph, scatter
    g = torch.Generator().manual_seed(seed)
    dataset = []
    for _ in range(num_samples):
        n_heavy = int(torch.randint(2, 9, (1,), generator=g))
        n_h = int(torch.randint(1, 2 * n_heavy, (1,), generator=g))
        z = torch.cat([
            torch.randint(6, 9, (n_heavy,), generator=g),
            torch.ones(n_h, dtype=torch.long)])
        n = z.numel()
        pos = torch.randn(n, 3, generator=g) * 1.5
        ei = radius_graph(pos, 4.0, max_num_neighbors=20)
        u = z.float().view(-1, 1) / 9.0
        nbr = scatter(u[ei[0]], ei[1], n, "mean")
        node_t = 0.5 * u + 0.5 * nbr
        graph_t = node_t.mean().view(1, 1)
        y = torch.cat([graph_t.view(-1), node_t.view(-1)]).view(-1, 1)
        d = Data(x=u, z=z, pos=pos, edge_index=ei, y=y,
                 y_loc=torch.tensor([[0, 1, 1 + n]], dtype=torch.long))
        d.num_nodes = n
        dataset.append(d)
    return dataset


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--mpnn_type", default=None)
    parser.add_argument("--num_epoch", type=int, default=None)
    parser.add_argument("--num_samples", type=int, default=200)
    args = parser.parse_args()

    with open(os.path.join(os.path.dirname(__file__), "qm9.json")) as f:
        config = json.load(f)
    if args.mpnn_type:
        config["NeuralNetwork"]["Architecture"]["mpnn_type"] = args.mpnn_type
    if args.num_epoch:
        config["NeuralNetwork"]["Training"]["num_epoch"] = args.num_epoch

    setup_ddp()
    torch.manual_seed(5)
    dataset = build_dataset(args.num_samples)
    trainset, valset, testset = split_dataset(
        dataset, config["NeuralNetwork"]["Training"]["perc_train"])
    train_loader, val_loader, test_loader = create_dataloaders(
        trainset, valset, testset,
        config["NeuralNetwork"]["Training"]["batch_size"], config=config)
    config = update_config(config, train_loader, val_loader, test_loader)

    log_name = get_log_name_config(config)
    setup_log(log_name)
    save_config(config, log_name)

    model = create_model_config(config["NeuralNetwork"],
                                verbosity=config["Verbosity"]["level"])
    model = distributed_model_wrapper(model)
    optimizer = select_optimizer(
        model, config["NeuralNetwork"]["Training"]["Optimizer"])
    scheduler = torch.optim.lr_scheduler.ReduceLROnPlateau(
        optimizer, mode="min", factor=0.5, patience=5)
    writer = get_summary_writer(log_name)

    train_validate_test(model, optimizer, train_loader, val_loader,
                        test_loader, writer, scheduler,
                        config["NeuralNetwork"], log_name,
                        config["Verbosity"]["level"],
                        create_plots=config.get("Visualization", {})
                        .get("create_plots", False))
    save_model(model, optimizer, log_name)


if __name__ == "__main__":
    main()
