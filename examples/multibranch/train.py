"""Multi-branch task-parallel training driver (reference
examples/multibranch/train.py:230-546): ranks are colored by branch,
each color trains its own decoder over a branch process group while the
shared encoder syncs over WORLD (MultiTaskModelMP).

Single process: runs both branches' data through the multi-branch model
(dataset_name masking).  Multi process (torchrun --nproc-per-node N):
rank r takes branch r % nbranch.
"""

import argparse
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from hydragnn_amd.models import create_model_config
from hydragnn_amd.models.multitask_mp import MultiTaskModelMP
from hydragnn_amd.preprocess import create_dataloaders
from hydragnn_amd.utils.config import update_config
from hydragnn_amd.utils.distributed import setup_ddp
from hydragnn_amd.ops import radius_graph, scatter


def make_branch_dataset(branch, num_samples=64, seed=3):
    from hydragnn_amd.data import Data
    g = torch.Generator().manual_seed(seed + branch)
    ds = []
    for _ in range(num_samples):
        n = 12
        pos = torch.rand(n, 3, generator=g) * 2
        ei = radius_graph(pos, 1.0, max_num_neighbors=20)
        u = torch.rand(n, 1, generator=g)
        nbr = scatter(u[ei[0]], ei[1], n, "mean")
        # branch-specific closed-form targets
        t = nbr.mean() if branch == 0 else (nbr ** 2).mean()
        d = Data(x=u, pos=pos, edge_index=ei, y=t.view(1, 1),
                 dataset_name=torch.tensor([[branch]]))
        d.num_nodes = n
        ds.append(d)
    return ds


CONFIG = {
    "Verbosity": {"level": 0},
    "Dataset": {"name": "multibranch_synthetic"},
    "NeuralNetwork": {
        "Architecture": {
            "mpnn_type": "GIN",
            "radius": 1.0,
            "max_neighbours": 20,
            "hidden_dim": 32,
            "num_conv_layers": 2,
            "output_heads": {"graph": [
                {"type": "branch-0", "architecture": {
                    "num_sharedlayers": 1, "dim_sharedlayers": 16,
                    "num_headlayers": 2, "dim_headlayers": [16, 16]}},
                {"type": "branch-1", "architecture": {
                    "num_sharedlayers": 1, "dim_sharedlayers": 16,
                    "num_headlayers": 2, "dim_headlayers": [16, 16]}},
            ]},
            "task_weights": [1.0],
        },
        "Variables_of_interest": {
            "input_node_features": [0],
            "output_names": ["target"], "output_index": [0],
            "output_dim": [1], "type": ["graph"],
            "denormalize_output": False,
        },
        "Training": {
            "num_epoch": 10, "perc_train": 0.8, "batch_size": 16,
            "loss_function_type": "mse",
            "Optimizer": {"type": "AdamW", "learning_rate": float(
                os.environ.get("HYDRAGNN_HPO_LR", 0.005))},
        },
    },
}


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--num_epoch", type=int, default=10)
    args = parser.parse_args()
    world_size, rank = setup_ddp()
    nbranch = 2
    mycolor = rank % nbranch

    branch_group = None
    if dist.is_initialized() and world_size > 1:
        for color in range(nbranch):
            ranks = [r for r in range(world_size) if r % nbranch == color]
            grp = dist.new_group(ranks=ranks)
            if color == mycolor:
                branch_group = grp

    config = dict(CONFIG)
    dataset = make_branch_dataset(mycolor)
    loaders = create_dataloaders(dataset, dataset, dataset, 16,
                                 config=config)
    config = update_config(config, *loaders)
    base = create_model_config(config["NeuralNetwork"], use_gpu=False)
    model = MultiTaskModelMP(base, mycolor, branch_group)
    opt = model.make_dual_optimizer(lr=0.005)

    for epoch in range(args.num_epoch):
        tot, nb = 0.0, 0
        for batch in loaders[0]:
            opt.zero_grad()
            pred = model(batch)
            head_index = [torch.arange(batch.y.shape[0])]
            loss, _ = model.loss(pred, batch.y, head_index)
            loss.backward()
            opt.step()
            tot += float(loss)
            nb += 1
        if rank == 0:
            print(f"epoch {epoch} branch {mycolor} loss {tot / nb:.6f}")


if __name__ == "__main__":
    main()
