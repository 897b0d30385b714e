"""Shared machinery for the example drivers.

The reference examples (reference examples/*) each download a public
dataset, preprocess it into an ADIOS store, and train through the
unified config pipeline.  This image has no network, so every example
here generates synthetic data OF THE SAME SHAPE (atom counts, target
layout, head structure) with closed-form learnable targets, and runs
the same config-driven flow: create_dataloaders -> update_config ->
create_model_config -> distributed_model_wrapper ->
train_validate_test.  Real datasets drop in through
utils/datasets (GraphStore, raw readers, or the ADIOS2 .bp converter,
utils/datasets/adios_reader.py).
"""

from __future__ import annotations

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from hydragnn_amd.data import Data                                # noqa: E402
from hydragnn_amd.models import create_model_config               # noqa: E402
from hydragnn_amd.ops import radius_graph, scatter                # noqa: E402
from hydragnn_amd.preprocess import (                             # noqa: E402
    create_dataloaders, split_dataset)
from hydragnn_amd.train import train_validate_test                # noqa: E402
from hydragnn_amd.utils.config import update_config               # noqa: E402
from hydragnn_amd.utils.datasets.synthetic import (               # noqa: E402
    _lj_energy_forces, lj_dataset)
from hydragnn_amd.utils.distributed import (                      # noqa: E402
    distributed_model_wrapper, setup_ddp)
from hydragnn_amd.utils.optimizer import select_optimizer         # noqa: E402


# ---------------------------------------------------------------------------
# generators
# ---------------------------------------------------------------------------
def mlip_molecules(num_samples, n_range=(8, 24), radius=5.0, seed=0,
                   species=(1, 6, 7, 8), spread_per_atom=0.45,
                   min_dist=0.9):
    """Variable-size molecules with LJ energies/forces (the ani1x /
    transition1x / OMol shape: MLIP targets over mixed species and
    sizes)."""
    g = torch.Generator().manual_seed(seed)
    ds = []
    for _ in range(num_samples):
        n = int(torch.randint(n_range[0], n_range[1] + 1, (1,),
                              generator=g))
        spread = spread_per_atom * n ** (1 / 3) * 2.0
        pos = (torch.rand(n, 3, generator=g) - 0.5) * 2 * spread
        for _ in range(40):
            d = torch.cdist(pos, pos) + torch.eye(n) * 10
            if float(d.min()) > min_dist:
                break
            pos = pos * 1.2
        z = torch.tensor(species, dtype=torch.long)[
            torch.randint(0, len(species), (n,), generator=g)]
        ei = radius_graph(pos, radius, max_num_neighbors=50)
        e, f = _lj_energy_forces(pos.double(), ei, None, 0.05, 1.0)
        d = Data(x=z.float().view(-1, 1), z=z, pos=pos, edge_index=ei,
                 energy=e.float().view(1, 1), forces=f.float(),
                 y=(e.float() / n).view(1, 1))
        d.num_nodes = n
        ds.append(d)
    return ds


def multihead_molecules(num_samples, graph_dims=(1,), node_dims=(),
                        n_range=(10, 18), radius=4.0, seed=0,
                        feat_dim=1):
    """Molecules with the reference's concatenated-y multihead layout:
    y = [graph targets..., node targets...] with y_loc offsets
    (reference train_validate_test.py:523).  Targets are closed-form
    functions of 1-hop neighbor means, so every head is learnable."""
    g = torch.Generator().manual_seed(seed)
    ds = []
    for _ in range(num_samples):
        n = int(torch.randint(n_range[0], n_range[1] + 1, (1,),
                              generator=g))
        pos = (torch.rand(n, 3, generator=g) - 0.5) * 4.0
        u = torch.rand(n, feat_dim, generator=g)
        ei = radius_graph(pos, radius, max_num_neighbors=50)
        src, dst = ei[0], ei[1]
        nbr = (scatter(u[src], dst, n, "sum") + u) / (
            scatter(torch.ones(src.shape[0], 1), dst, n, "sum") + 1.0)
        y_parts, y_loc = [], [0]
        for k, dim in enumerate(graph_dims):
            t = torch.stack([nbr.mean() ** (j + 1)
                             for j in range(dim)]).view(-1)
            y_parts.append(t)
            y_loc.append(y_loc[-1] + dim)
        for k, dim in enumerate(node_dims):
            t = torch.cat([nbr[:, :1] ** (j + 1) for j in range(dim)],
                          dim=1)
            y_parts.append(t.reshape(-1))
            y_loc.append(y_loc[-1] + n * dim)
        d = Data(x=u, pos=pos, edge_index=ei,
                 y=torch.cat(y_parts).view(-1, 1),
                 y_loc=torch.tensor([y_loc], dtype=torch.long))
        d.num_nodes = n
        ds.append(d)
    return ds


def topology_graphs(num_samples, n_range=(10, 30), p=0.25, seed=0,
                    feat_dim=4, out_dim=1):
    """Positionless graphs (ZINC / OGB shape: molecular topology +
    node/edge features, no geometry).  Graph target = closed-form in
    degrees and features."""
    g = torch.Generator().manual_seed(seed)
    ds = []
    for _ in range(num_samples):
        n = int(torch.randint(n_range[0], n_range[1] + 1, (1,),
                              generator=g))
        adj = (torch.rand(n, n, generator=g) < p)
        adj = adj | adj.t()
        adj.fill_diagonal_(False)
        # keep connected-ish: chain backbone
        idx = torch.arange(n - 1)
        adj[idx, idx + 1] = True
        adj[idx + 1, idx] = True
        ei = adj.nonzero().t().contiguous()
        x = torch.rand(n, feat_dim, generator=g)
        deg = adj.sum(1).float()
        base = (x.mean() + deg.mean() / n)
        y = torch.stack([base ** (j + 1)
                         for j in range(out_dim)]).view(1, -1)
        d = Data(x=x, edge_index=ei, y=y.view(-1, 1),
                 edge_attr=torch.rand(ei.shape[1], 3, generator=g))
        d.num_nodes = n
        ds.append(d)
    return ds


def spectrum_molecules(num_samples, n_freq=64, n_range=(10, 16),
                       radius=4.0, seed=0, smooth=True):
    """UV-spectrum shape (reference examples/dftb_uv_spectrum): one
    graph head whose output is a WHOLE spectrum vector.  Synthetic
    spectrum = Gaussian (smooth) or nearest-bin (discrete) projection
    of the graph-Laplacian eigenvalues onto a frequency grid."""
    g = torch.Generator().manual_seed(seed)
    grid = torch.linspace(0.0, 8.0, n_freq)
    ds = []
    for _ in range(num_samples):
        n = int(torch.randint(n_range[0], n_range[1] + 1, (1,),
                              generator=g))
        pos = (torch.rand(n, 3, generator=g) - 0.5) * 4.0
        ei = radius_graph(pos, radius, max_num_neighbors=50)
        A = torch.zeros(n, n)
        A[ei[0], ei[1]] = 1.0
        L = torch.diag(A.sum(1)) - A
        ev = torch.linalg.eigvalsh(L)
        if smooth:
            spec = torch.exp(
                -(grid.view(1, -1) - ev.view(-1, 1)) ** 2 / 0.5
            ).sum(0)
        else:
            spec = torch.zeros(n_freq)
            bins = ((ev / 8.0 * (n_freq - 1)).round().long()
                    .clamp(0, n_freq - 1))
            for b in bins:
                spec[b] += 1.0
        u = torch.rand(n, 1, generator=g)
        d = Data(x=u, pos=pos, edge_index=ei,
                 y=(spec / n).view(-1, 1))
        d.num_nodes = n
        ds.append(d)
    return ds


# ---------------------------------------------------------------------------
# config + flow
# ---------------------------------------------------------------------------
def mlip_config(mpnn_type="MACE", radius=5.0, hidden_dim=32,
                num_conv_layers=2, lr=0.003, batch_size=8,
                num_epoch=6, force_weight=20.0, extra_arch=None):
    arch = {
        "mpnn_type": mpnn_type, "radius": radius, "max_neighbours": 50,
        "hidden_dim": hidden_dim, "num_conv_layers": num_conv_layers,
        "enable_interatomic_potential": True,
        "energy_weight": 1.0, "energy_peratom_weight": 1.0,
        "force_weight": force_weight,
        "output_heads": {"node": {"num_headlayers": 2,
                                  "dim_headlayers": [hidden_dim,
                                                     hidden_dim],
                                  "type": "mlp"}},
        "task_weights": [1.0],
    }
    if mpnn_type == "MACE":
        arch.update({"max_ell": 2, "node_max_ell": 1, "correlation": 2,
                     "num_radial": 8, "equivariance": True})
    if extra_arch:
        arch.update(extra_arch)
    return {
        "Verbosity": {"level": 0},
        "Dataset": {"name": "synthetic"},
        "NeuralNetwork": {
            "Architecture": arch,
            "Variables_of_interest": {
                "input_node_features": [0],
                "output_names": ["energy"], "output_index": [0],
                "output_dim": [1], "type": ["node"],
                "denormalize_output": False,
            },
            "Training": {
                "num_epoch": num_epoch, "perc_train": 0.8,
                "batch_size": batch_size,
                "loss_function_type": "mse", "EarlyStopping": False,
                "Checkpoint": False,
                "Optimizer": {"type": "AdamW", "learning_rate": lr},
            },
        },
    }


def multihead_config(mpnn_type, head_types, output_dims,
                     hidden_dim=32, num_conv_layers=2, radius=4.0,
                     lr=0.005, batch_size=8, num_epoch=6,
                     output_names=None, extra_arch=None,
                     input_features=1):
    output_heads = {}
    if "graph" in head_types:
        output_heads["graph"] = {
            "num_sharedlayers": 1, "dim_sharedlayers": hidden_dim,
            "num_headlayers": 2,
            "dim_headlayers": [hidden_dim, hidden_dim]}
    if "node" in head_types:
        output_heads["node"] = {
            "num_headlayers": 2,
            "dim_headlayers": [hidden_dim, hidden_dim], "type": "mlp"}
    arch = {
        "mpnn_type": mpnn_type, "radius": radius, "max_neighbours": 50,
        "hidden_dim": hidden_dim, "num_conv_layers": num_conv_layers,
        "output_heads": output_heads,
        "task_weights": [1.0] * len(head_types),
    }
    if extra_arch:
        arch.update(extra_arch)
    return {
        "Verbosity": {"level": 0},
        "Dataset": {"name": "synthetic"},
        "NeuralNetwork": {
            "Architecture": arch,
            "Variables_of_interest": {
                "input_node_features": list(range(input_features)),
                "output_names": output_names
                or [f"t{i}" for i in range(len(head_types))],
                "output_index": list(range(len(head_types))),
                "output_dim": list(output_dims),
                "type": list(head_types),
                "denormalize_output": False,
            },
            "Training": {
                "num_epoch": num_epoch, "perc_train": 0.8,
                "batch_size": batch_size,
                "loss_function_type": "mse", "EarlyStopping": False,
                "Checkpoint": False,
                "Optimizer": {"type": "AdamW", "learning_rate": lr},
            },
        },
    }


def standard_args(num_samples_default=32):
    p = argparse.ArgumentParser()
    p.add_argument("--num_epoch", type=int, default=None)
    p.add_argument("--num_samples", type=int,
                   default=num_samples_default)
    return p.parse_args()


def run_flow(config, dataset, log_name, num_epoch=None, seed=17):
    """The standard example body (mirrors reference examples' unified
    flow)."""
    if num_epoch:
        config["NeuralNetwork"]["Training"]["num_epoch"] = num_epoch
    setup_ddp()
    torch.manual_seed(seed)
    bs = config["NeuralNetwork"]["Training"]["batch_size"]
    splits = split_dataset(
        dataset, config["NeuralNetwork"]["Training"]["perc_train"])
    loaders = create_dataloaders(*splits, bs, config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"])
    model = distributed_model_wrapper(model)
    opt = select_optimizer(
        model, config["NeuralNetwork"]["Training"]["Optimizer"])
    train_validate_test(model, opt, *loaders, writer=None,
                        scheduler=None,
                        config=config["NeuralNetwork"],
                        log_name=log_name, verbosity=0)
    return model, config
