"""Deterministic synthetic graph data with closed-form targets.

Same test strategy as the reference's
tests/deterministic_graph_data.py:20-173 (lattice graphs whose node
targets are neighborhood averages so message passing is required, graph
target = their normalized sum), so end-to-end training can be asserted
against absolute error thresholds.
"""

from __future__ import annotations

import math

import torch

from hydragnn_amd.data import Data
from hydragnn_amd.ops import radius_graph, scatter, gather


def make_deterministic_dataset(
    num_samples: int = 64,
    lattice: int = 3,
    radius: float = 1.2,
    num_heads_node: int = 1,
    include_graph_head: bool = True,
    seed: int = 7,
):
    """Each sample: lattice^3 nodes on a unit grid; node scalar feature
    u in [0,1]; node target = 1-hop mean of u (incl. self); graph
    target = mean of node targets.  data.y is the concatenated
    [graph_dims..., node_dims*N...] layout with y_loc offsets, matching
    the reference convention (train_validate_test.py:523)."""
    g = torch.Generator().manual_seed(seed)
    n = lattice ** 3
    grid = torch.stack(torch.meshgrid(
        torch.arange(lattice), torch.arange(lattice), torch.arange(lattice),
        indexing="ij"), dim=-1).reshape(-1, 3).float()

    num_heads = (1 if include_graph_head else 0) + num_heads_node
    dataset = []
    for _ in range(num_samples):
        u = torch.rand(n, 1, generator=g)
        pos = grid.clone()
        edge_index = radius_graph(pos, radius, max_num_neighbors=100)
        src, dst = edge_index[0], edge_index[1]
        nbr_sum = scatter(u[src], dst, n, "sum") + u
        deg = scatter(torch.ones(src.shape[0], 1), dst, n, "sum") + 1.0
        t = nbr_sum / deg  # node target
        graph_t = t.mean().view(1, 1)

        y_parts = []
        y_loc = [0]
        if include_graph_head:
            y_parts.append(graph_t.view(-1))
            y_loc.append(y_loc[-1] + 1)
        for h in range(num_heads_node):
            tgt = t if h == 0 else t ** (h + 1)
            y_parts.append(tgt.view(-1))
            y_loc.append(y_loc[-1] + n)
        y = torch.cat(y_parts).view(-1, 1)
        data = Data(
            x=u,
            pos=pos,
            edge_index=edge_index,
            y=y,
            y_loc=torch.tensor([y_loc], dtype=torch.long),
        )
        data.num_nodes = n
        dataset.append(data)
    return dataset


def base_config(mpnn_type: str = "GIN", heads=("graph",),
                num_epoch: int = 40, hidden_dim: int = 16,
                num_conv_layers: int = 2, lr: float = 0.02,
                batch_size: int = 16):
    output_heads = {}
    head_types = []
    if "graph" in heads:
        output_heads["graph"] = {
            "num_sharedlayers": 2,
            "dim_sharedlayers": 16,
            "num_headlayers": 2,
            "dim_headlayers": [16, 16],
        }
        head_types.append("graph")
    if "node" in heads:
        output_heads["node"] = {
            "num_headlayers": 2,
            "dim_headlayers": [16, 16],
            "type": "mlp",
        }
        head_types.append("node")
    return {
        "Verbosity": {"level": 0},
        "Dataset": {"name": "unit_test"},
        "NeuralNetwork": {
            "Architecture": {
                "mpnn_type": mpnn_type,
                "radius": 1.2,
                "max_neighbours": 100,
                "hidden_dim": hidden_dim,
                "num_conv_layers": num_conv_layers,
                "output_heads": output_heads,
                "task_weights": [1.0] * len(head_types),
            },
            "Variables_of_interest": {
                "input_node_features": [0],
                "output_index": list(range(len(head_types))),
                "type": head_types,
                "denormalize_output": False,
            },
            "Training": {
                "num_epoch": num_epoch,
                "perc_train": 0.7,
                "EarlyStopping": False,
                "patience": 10,
                "Checkpoint": False,
                "checkpoint_warmup": 10,
                "loss_function_type": "mse",
                "batch_size": batch_size,
                "Optimizer": {
                    "type": "AdamW",
                    "use_zero_redundancy": False,
                    "learning_rate": lr,
                },
            },
        },
        "Visualization": {"create_plots": False},
    }
