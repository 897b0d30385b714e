"""Secondary benchmark configs (BASELINE.json configs[2..4]): PaiNN
multi-head QM9-shape, EGNN + GPS global attention, DimeNet fp64 with
gradient checkpointing on periodic cells.  The driver's contract bench
(bench.py) stays MACE; this script produces the per-config evidence
lines.

Usage: python scripts/bench_configs.py --config painn_qm9 --steps 10
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from hydragnn_amd.data import Batch
from hydragnn_amd.models.create import create_model, resolve_precision
from hydragnn_amd.train import get_autocast_and_scaler, get_head_indices
from hydragnn_amd.utils.datasets.synthetic import lj_dataset


def qm9_shape_multihead(num_samples, seed=23):
    """QM9-shaped molecules with graph + node targets + Laplacian PE."""
    from hydragnn_amd.data import Data
    from hydragnn_amd.ops import radius_graph, scatter
    g = torch.Generator().manual_seed(seed)
    ds = []
    for _ in range(num_samples):
        n = 18
        pos = torch.randn(n, 3, generator=g) * 1.5
        ei = radius_graph(pos, 4.0, max_num_neighbors=20)
        u = torch.rand(n, 1, generator=g)
        nbr = scatter(u[ei[0]], ei[1], n, "mean")
        y = torch.cat([nbr.mean().view(1), nbr.view(-1)]).view(-1, 1)
        d = Data(x=u, pos=pos, edge_index=ei, y=y,
                 y_loc=torch.tensor([[0, 1, 1 + n]]))
        d.num_nodes = n
        ds.append(d)
    return ds


def oc20_shape(num_samples, seed=29, with_pe=False):
    """OC20-shaped surfaces: ~80 atoms, periodic slab."""
    from hydragnn_amd.preprocess import add_laplacian_pe
    ds = lj_dataset(num_samples=num_samples, num_atoms=80,
                    cell_size=12.0, radius=4.0, pbc=True, seed=seed)
    if with_pe:
        for d in ds:
            add_laplacian_pe(d, 3)
    return ds


CONFIGS = {
    "painn_qm9": dict(
        label="PaiNN multi-head (graph+node) QM9-shape",
        precision="bf16", local_batch=256, mlip=False,
        data=lambda b: qm9_shape_multihead(b),
        model=dict(mpnn_type="PAINN", input_dim=1, hidden_dim=64,
                   output_dim=[1, 1], output_type=["graph", "node"],
                   output_heads={
                       "graph": [{"type": "branch-0", "architecture": {
                           "num_sharedlayers": 1, "dim_sharedlayers": 64,
                           "num_headlayers": 2,
                           "dim_headlayers": [64, 64]}}],
                       "node": [{"type": "branch-0", "architecture": {
                           "num_headlayers": 2,
                           "dim_headlayers": [64, 64],
                           "type": "mlp"}}]},
                   task_weights=[1.0, 1.0], num_conv_layers=3,
                   radius=4.0, num_radial=20, max_neighbours=20,
                   equivariance=True, activation_function="silu")),
    "egnn_gps": dict(
        label="EGNN + GPS global attention, OC20-shape, bf16",
        precision="bf16", local_batch=32, mlip=False,
        data=lambda b: oc20_shape(b, with_pe=True),
        model=dict(mpnn_type="EGNN", input_dim=1, hidden_dim=64,
                   output_dim=[1], output_type=["graph"],
                   output_heads={"graph": [{
                       "type": "branch-0", "architecture": {
                           "num_sharedlayers": 1, "dim_sharedlayers": 64,
                           "num_headlayers": 2,
                           "dim_headlayers": [64, 64]}}]},
                   task_weights=[1.0], num_conv_layers=3, radius=4.0,
                   max_neighbours=30, equivariance=True, pe_dim=3,
                   global_attn_engine="gps",
                   global_attn_type="multihead", global_attn_heads=4,
                   activation_function="silu")),
    "dimenet_fp64": dict(
        label="DimeNet fp64 + grad checkpointing, periodic cells",
        precision="fp64", local_batch=4, mlip=False,
        # 1024-atom FeSi-shape cells at the 216-atom run's density
        # (12^3 for 216 -> 20.25^3 for 1024); BASELINE configs[4]
        data=lambda b: lj_dataset(num_samples=b, num_atoms=1024,
                                  cell_size=20.25, radius=3.0,
                                  pbc=True, seed=31,
                                  dtype=torch.float64),
        model=dict(mpnn_type="DimeNet", input_dim=1, hidden_dim=64,
                   output_dim=[1], output_type=["graph"],
                   output_heads={"graph": [{
                       "type": "branch-0", "architecture": {
                           "num_sharedlayers": 1, "dim_sharedlayers": 64,
                           "num_headlayers": 2,
                           "dim_headlayers": [64, 64]}}]},
                   task_weights=[1.0], num_conv_layers=2, radius=3.0,
                   num_radial=6, num_spherical=7, basis_emb_size=8,
                   int_emb_size=64, out_emb_size=128,
                   num_before_skip=1, num_after_skip=2,
                   max_neighbours=30, conv_checkpointing=True)),
}


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--config", default="painn_qm9",
                        choices=sorted(CONFIGS))
    parser.add_argument("--steps", type=int, default=10)
    parser.add_argument("--warmup", type=int, default=3)
    parser.add_argument("--batch", type=int, default=None)
    args = parser.parse_args()
    cfg = CONFIGS[args.config]
    use_cuda = torch.cuda.is_available()
    device = "cuda:0" if use_cuda else "cpu"
    torch.manual_seed(7)

    batch_size = args.batch or cfg["local_batch"]
    prec, param_dtype, _ = resolve_precision(cfg["precision"])
    model = create_model(use_gpu=False, loss_function_type="mse",
                         **cfg["model"])
    model = model.to(device=device, dtype=param_dtype)
    data = cfg["data"](batch_size)
    batch = Batch.from_data_list(data).to(device)
    for key in list(batch.keys()):
        v = batch[key]
        if torch.is_tensor(v) and torch.is_floating_point(v):
            batch[key] = v.to(param_dtype)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    autocast, _ = get_autocast_and_scaler(cfg["precision"])
    head_index = get_head_indices(model, batch)

    def step():
        opt.zero_grad(set_to_none=True)
        with autocast:
            pred = model(batch)
            loss, _ = model.loss(pred, batch.y, head_index)
        loss.backward()
        opt.step()
        return loss

    for _ in range(args.warmup):
        step()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if use_cuda:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({
        "metric": f"graphs/sec training ({cfg['label']})",
        "value": batch_size * args.steps / dt,
        "unit": "graphs/s",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": dt / args.steps * 1000,
        "higher_is_better": True,
        "dtype": cfg["precision"],
        "data": "synthetic",
        "config": {"model": cfg["label"], "global_batch": batch_size},
    }))


if __name__ == "__main__":
    main()
