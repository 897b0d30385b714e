"""Training / validation / test loop.

Behavioral parity with /root/reference/hydragnn/train/
train_validate_test.py:182-1124: per-epoch train/val/test with
multi-task loss aggregation, head-index bookkeeping over concatenated
data.y, precision plumbing (bf16 autocast / fp32 / fp64), checkpoint
trigger with warmup, early stopping, eval-only mode
(HYDRAGNN_EVALONLY), HYDRAGNN_MAX_NUM_BATCH clamp, distributed metric
reductions and padded eval gathers over RCCL.
"""

from __future__ import annotations

import os
from contextlib import nullcontext

import torch
import torch.distributed as dist

from ..models.create import resolve_precision
from ..utils.distributed import (check_remaining_time, get_device,
                                 is_fsdp2_enabled,
                                 set_reshard_after_backward)
from ..utils.model.model import Checkpoint, EarlyStopping, save_model
from ..utils.print.print_utils import iterate_tqdm, log, print_distributed
from ..utils.profiling_and_tracing import tracer as tr
from ..utils.profiling_and_tracing.profile import Profiler
from ..utils.profiling_and_tracing.time_utils import Timer


def _unwrap(model):
    return model.module if hasattr(model, "module") else model


def _model_device(model):
    """Device of the model's parameters — batches must follow the
    MODEL, not the globally preferred device (a CPU-built model on a
    GPU box would otherwise receive cuda batches and fail in
    F.linear)."""
    for p in model.parameters():
        return p.device
    return get_device()


def move_batch_to_device(data, param_dtype, device=None):
    if device is None:
        device = get_device()
    for key in list(data.keys()):
        v = data[key]
        if torch.is_tensor(v) and torch.is_floating_point(v):
            data[key] = v.to(dtype=param_dtype)
    return data.to(device, non_blocking=True)


def get_autocast_and_scaler(precision, device=None):
    """Autocast context + loss scaler per precision (reference
    train_validate_test.py:87-110 + GradScaler machinery :783-801):
    bf16 -> autocast, no scaler; fp16 -> autocast + GradScaler (loss
    scaling against underflow; GPU only); fp32/fp64 -> neither."""
    precision, _, autocast_dtype = resolve_precision(precision)
    if device is None:
        device = get_device()
    if precision == "bf16":
        use_bf16 = device.type == "cuda" or bool(
            getattr(torch.backends.cpu, "has_bf16", False))
        if use_bf16:
            return torch.autocast(device_type=device.type,
                                  dtype=autocast_dtype), None
    if precision == "fp16":
        if device.type == "cuda":
            scaler = torch.amp.GradScaler("cuda")
            return torch.autocast(device_type="cuda",
                                  dtype=autocast_dtype), scaler
        log("Requested fp16 on CPU; falling back to full precision.")
    return nullcontext(), None


def get_nbatch(loader):
    nbatch = len(loader)
    if os.getenv("HYDRAGNN_MAX_NUM_BATCH") is not None:
        nbatch = min(nbatch, int(os.environ["HYDRAGNN_MAX_NUM_BATCH"]))
    if dist.is_initialized() and dist.get_world_size() > 1:
        from ..utils.distributed import to_comm_device
        t, _ = to_comm_device(torch.tensor([nbatch], dtype=torch.long))
        dist.all_reduce(t, op=dist.ReduceOp.MIN)
        nbatch = int(t.item())
    return nbatch


# ---------------------------------------------------------------------------
# head indices over concatenated data.y
# ---------------------------------------------------------------------------
def get_head_indices(model, data):
    m = _unwrap(model)
    if all(t == "graph" for t in m.head_type):
        return _head_indices_graph(m, data)
    return _head_indices_node_or_mixed(m, data)


def _head_indices_graph(m, data):
    nsize = data.y.shape[0]
    if m.num_heads == 1:
        return [torch.arange(nsize, device=data.y.device)]
    head_dims = m.head_dims
    head_dimsum = sum(head_dims)
    batch_size = int(data.batch.max()) + 1
    head_index = []
    dev = data.y.device
    for ihead in range(m.num_heads):
        per = torch.arange(head_dims[ihead], device=dev).repeat(batch_size)
        shift = sum(head_dims[:ihead]) + torch.repeat_interleave(
            torch.arange(batch_size, device=dev) * head_dimsum,
            head_dims[ihead])
        head_index.append(per + shift)
    return head_index


def _head_indices_node_or_mixed(m, data):
    if m.num_heads == 1:
        return [torch.arange(data.y.shape[0], device=data.y.device)]
    y_loc = data.y_loc
    batch_size = int(data.batch.max()) + 1
    total_size = y_loc[:, -1]
    sample_start = (torch.cumsum(total_size, 0) - total_size).view(-1, 1)
    start_index = sample_start + y_loc[:, :-1]
    end_index = sample_start + y_loc[:, 1:]
    index_range = torch.arange(int(end_index[-1, -1]), device=y_loc.device)
    head_index = []
    for ihead in range(m.num_heads):
        segs = [index_range[int(start_index[s, ihead]):
                            int(end_index[s, ihead])]
                for s in range(batch_size)]
        head_index.append(torch.cat(segs, 0))
    return head_index


# ---------------------------------------------------------------------------
# metric reductions
# ---------------------------------------------------------------------------
@torch.no_grad()
def reduce_values_ranks(local_tensor, count=None):
    """Cross-rank mean; HYDRAGNN_AGGR_BACKEND=mpi routes the tiny
    metric reductions through MPI instead of RCCL (the reference's
    at-scale tuning) when mpi4py is available.

    With `count` (local sample count), `local_tensor` is a local SUM and
    the global mean is sum-of-sums / sum-of-counts — unbiased when ranks
    hold unequal sample counts (node-budget batching, uneven shards).
    Without it, falls back to averaging per-rank values equally."""
    if not (dist.is_initialized() and dist.get_world_size() > 1):
        if count is not None:
            return local_tensor / max(float(count), 1.0)
        return local_tensor
    if count is not None:
        packed = torch.cat([
            local_tensor.reshape(-1).double(),
            torch.tensor([float(count)], dtype=torch.float64,
                         device=local_tensor.device)])
        packed = _allreduce_sum(packed)
        total = packed[:-1] / packed[-1].clamp(min=1.0)
        return total.to(local_tensor.dtype).view_as(local_tensor)
    return _allreduce_sum(local_tensor) / dist.get_world_size()


@torch.no_grad()
def _allreduce_sum(t):
    if os.getenv("HYDRAGNN_AGGR_BACKEND") == "mpi":
        try:
            from mpi4py import MPI
            out = MPI.COMM_WORLD.allreduce(
                t.detach().cpu().numpy(), op=MPI.SUM)
            return torch.as_tensor(out, dtype=t.dtype, device=t.device)
        except ImportError:
            pass
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


@torch.no_grad()
def gather_tensor_ranks(head_values):
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return head_values
    orig_device = head_values.device
    # gather on the device the backend communicates on: RCCL ("nccl")
    # needs cuda tensors, gloo needs cpu (ROCm gloo has no CUDA-tensor
    # support)
    if dist.get_backend() == "nccl":
        from ..utils.distributed import to_comm_device
        head_values, _ = to_comm_device(head_values)
    elif head_values.is_cuda:
        head_values = head_values.cpu()
    size_local = torch.tensor([head_values.shape[0]], dtype=torch.int64,
                              device=head_values.device)
    size_all = [torch.ones_like(size_local)
                for _ in range(dist.get_world_size())]
    dist.all_gather(size_all, size_local)
    size_all = torch.cat(size_all, 0)
    max_size = int(size_all.max())
    padded = torch.zeros(max_size, *head_values.shape[1:],
                         dtype=head_values.dtype, device=head_values.device)
    padded[: head_values.shape[0]] = head_values
    gathered = [torch.zeros_like(padded)
                for _ in range(dist.get_world_size())]
    dist.all_gather(gathered, padded)
    return torch.cat([g[: int(s)] for g, s in zip(gathered, size_all)],
                     0).to(orig_device)


# ---------------------------------------------------------------------------
# inner loops
# ---------------------------------------------------------------------------
def _compute_loss(model, data, use_interatomic: bool, create_graph: bool):
    m = _unwrap(model)
    if use_interatomic:
        data.pos.requires_grad_(True)
        pred = model(data)
        loss, tasks_loss = m.energy_force_loss(
            pred, data, create_graph=create_graph)
    else:
        head_index = get_head_indices(model, data)
        pred = model(data)
        loss, tasks_loss = m.loss(pred, data.y, head_index)
    return loss, tasks_loss, pred


def train(loader, model, opt, verbosity, precision="fp32", profiler=None):
    if profiler is None:
        profiler = Profiler()
    m = _unwrap(model)
    num_tasks = len(m.loss_weights) if not _use_interatomic(m) else 3
    device = _model_device(m)
    total_error = torch.zeros(1, device=device)
    tasks_error = torch.zeros(num_tasks, device=device)
    num_samples_local = 0
    model.train()
    _, param_dtype, _ = resolve_precision(precision)
    autocast, scaler = get_autocast_and_scaler(precision, device)
    nbatch = get_nbatch(loader)
    use_ia = _use_interatomic(m)
    dataset = getattr(loader, "dataset", None)
    if dataset is not None and hasattr(dataset, "epoch_begin"):
        dataset.epoch_begin()   # DDStore-style fetch window
    if use_ia and is_fsdp2_enabled():
        # FSDP2 double-backward workaround (see set_reshard_after_backward)
        set_reshard_after_backward(model, False)

    from .captured import capture_enabled, get_or_build_stepper
    from ..utils.distributed import is_deepspeed_engine
    ds_engine = is_deepspeed_engine(model)
    stepper = None

    for ibatch, data in enumerate(iterate_tqdm(loader, verbosity)):
        if ibatch >= nbatch:
            break
        from ..globalatt.gps import redraw_performer_projections
        redraw_performer_projections(
            model, int(os.environ.get("HYDRAGNN_ATTN_REDRAW", "1000")))

        # hipGraph-captured fast path (static-shape batches on GPU):
        # H2D copy-in + one graph replay per batch (captured.py).
        if (scaler is None and stepper is None and not ds_engine
                and capture_enabled(data, model)):
            dev_data = move_batch_to_device(data, param_dtype, device)
            # capture the UNWRAPPED module's forward: DDP's python
            # pre/post-forward stays entirely out of the graph (its
            # reducer is unused on this path — FlatGradSync owns the
            # gradient sync), removing the main capture-failure risk
            # at world > 1
            stepper = get_or_build_stepper(
                model, opt, dev_data, autocast,
                lambda b: _compute_loss(_unwrap(model), b, use_ia,
                                        create_graph=True),
                param_dtype)
            if stepper is not None:
                data = dev_data
        if stepper is not None and stepper.matches(data):
            tr.start("captured_step")
            loss, tasks_loss = stepper.step(data)
            tr.stop("captured_step")
        else:
            tr.start("h2d")
            data = move_batch_to_device(data, param_dtype, device)
            tr.stop("h2d")
            if not ds_engine:
                opt.zero_grad(set_to_none=True)
            tr.start("forward")
            with autocast:
                loss, tasks_loss, _ = _compute_loss(model, data, use_ia,
                                                    create_graph=True)
            tr.stop("forward")
            tr.start("backward")
            if ds_engine:
                # DeepSpeed engine owns loss scaling / grad
                # accumulation (reference train_validate_test.py:729)
                model.backward(loss)
            elif scaler is not None:
                scaler.scale(loss).backward()
            else:
                loss.backward()
            tr.stop("backward")
            tr.start("opt_step")
            clip = getattr(opt, "_hydragnn_grad_clip", None)
            if clip is not None and scaler is None and not ds_engine:
                torch.nn.utils.clip_grad_norm_(
                    (p for p in model.parameters()
                     if p.grad is not None), clip)
            if ds_engine:
                model.step()
            elif scaler is not None:
                scaler.step(opt)
                scaler.update()
            else:
                opt.step()
            tr.stop("opt_step")
        if use_ia and is_fsdp2_enabled():
            set_reshard_after_backward(model, True)
            set_reshard_after_backward(model, False)
        profiler.step()
        n = int(data.get("num_real_graphs_", 0) or data.num_graphs) \
            if hasattr(data, "get") else data.num_graphs
        total_error += loss.detach() * n
        for it, tl in enumerate(tasks_loss):
            tasks_error[it] += tl.detach() * n
        num_samples_local += n

    if dataset is not None and hasattr(dataset, "epoch_end"):
        dataset.epoch_end()
    train_error = reduce_values_ranks(total_error, num_samples_local)
    train_tasks_error = reduce_values_ranks(tasks_error,
                                            num_samples_local)
    return train_error, train_tasks_error


@torch.no_grad()
def _eval_pass(loader, model, verbosity, precision, return_samples=False):
    m = _unwrap(model)
    use_ia = _use_interatomic(m)
    # DDStore-style fetch window for sharded/remote datasets
    _ds = getattr(loader, "dataset", None)
    _window = _ds is not None and hasattr(_ds, "epoch_begin")
    if _window:
        _ds.epoch_begin()
    num_tasks = len(m.loss_weights) if not use_ia else 3
    device = _model_device(m)
    total_error = torch.zeros(1, device=device)
    tasks_error = torch.zeros(num_tasks, device=device)
    num_samples_local = 0
    model.eval()
    _, param_dtype, _ = resolve_precision(precision)
    autocast, _ = get_autocast_and_scaler(precision, device)
    nbatch = get_nbatch(loader)
    true_values = [[] for _ in range(m.num_heads)]
    predicted_values = [[] for _ in range(m.num_heads)]

    for ibatch, data in enumerate(loader):
        if ibatch >= nbatch:
            break
        data = move_batch_to_device(data, param_dtype, device)
        if use_ia:
            with torch.enable_grad():
                data.pos.requires_grad_(True)
                with autocast:
                    pred = model(data)
                    loss, tasks_loss = m.energy_force_loss(
                        pred, data, create_graph=False)
        else:
            head_index = get_head_indices(model, data)
            with autocast:
                pred = model(data)
                loss, tasks_loss = m.loss(pred, data.y, head_index)
            if return_samples:
                ytrue = data.y
                for ihead in range(m.num_heads):
                    p = pred[0][ihead] if m.var_output else pred[ihead]
                    true_values[ihead].append(
                        ytrue[head_index[ihead]].reshape(-1, 1).detach())
                    predicted_values[ihead].append(
                        p.reshape(-1, 1).detach())
        n = data.num_graphs
        total_error += loss.detach() * n
        for it, tl in enumerate(tasks_loss):
            tasks_error[it] += tl.detach() * n
        num_samples_local += n

    if _window:
        _ds.epoch_end()
    err = reduce_values_ranks(total_error, num_samples_local)
    tasks_err = reduce_values_ranks(tasks_error, num_samples_local)
    if return_samples:
        tv = [gather_tensor_ranks(torch.cat(v, 0)) if v else torch.zeros(0, 1)
              for v in true_values]
        pv = [gather_tensor_ranks(torch.cat(v, 0)) if v else torch.zeros(0, 1)
              for v in predicted_values]
        return err, tasks_err, tv, pv
    return err, tasks_err


def validate(loader, model, verbosity, precision="fp32"):
    return _eval_pass(loader, model, verbosity, precision)


def test(loader, model, verbosity, precision="fp32", return_samples=True):
    out = _eval_pass(loader, model, verbosity, precision,
                     return_samples=return_samples)
    if return_samples:
        return out
    err, tasks_err = out
    return err, tasks_err, [], []


def _use_interatomic(m) -> bool:
    return hasattr(m, "energy_force_loss")


# ---------------------------------------------------------------------------
# outer loop
# ---------------------------------------------------------------------------
def train_validate_test(
    model,
    optimizer,
    train_loader,
    val_loader,
    test_loader,
    writer,
    scheduler,
    config,
    log_name,
    verbosity,
    create_plots=False,
    compute_grad_energy=False,
):
    num_epoch = config["Training"]["num_epoch"]
    epoch_start = config["Training"].get("epoch_start", 0)
    precision = config["Training"].get("precision", "fp32")
    EarlyStop = config["Training"].get("EarlyStopping", False)
    early_stopping = EarlyStopping(
        patience=config["Training"].get("patience", 10)) if EarlyStop else None
    use_checkpoint = config["Training"].get("Checkpoint", False)
    checkpoint = Checkpoint(
        name=log_name,
        warmup=config["Training"].get("checkpoint_warmup", 0),
    ) if use_checkpoint else None

    profiler = Profiler(config.get("Profile", {}))
    timer = Timer("train_validate_test")
    timer.start()

    if os.getenv("HYDRAGNN_EVALONLY"):
        err, tasks_err, tv, pv = test(test_loader, model, verbosity,
                                      precision)
        print_distributed(verbosity, f"eval-only test error: {err}")
        return

    total_epochs = 0
    import time as _time
    for epoch in range(epoch_start, num_epoch):
        epoch_t0 = _time.time()
        for loader in (train_loader, val_loader, test_loader):
            sampler = getattr(loader, "sampler", None)
            if sampler is not None and hasattr(sampler, "set_epoch"):
                sampler.set_epoch(epoch)
            bsampler = getattr(loader, "batch_sampler", None)
            if bsampler is not None and hasattr(bsampler, "set_epoch"):
                bsampler.set_epoch(epoch)

        profiler.set_epoch(epoch)
        tr.start("train")
        train_error, train_tasks_error = train(
            train_loader, model, optimizer, verbosity, precision, profiler)
        tr.stop("train")
        if os.getenv("HYDRAGNN_VALTEST", "1") != "0":
            val_error, val_tasks_error = validate(
                val_loader, model, verbosity, precision)
            test_error, test_tasks_error = _eval_pass(
                test_loader, model, verbosity, precision)
        else:
            val_error = train_error.clone()
            test_error = train_error.clone()

        if scheduler is not None:
            scheduler.step(val_error)
        if writer is not None:
            writer.add_scalar("train_error", train_error.item(), epoch)
            writer.add_scalar("val_error", val_error.item(), epoch)
            writer.add_scalar("test_error", test_error.item(), epoch)

        print_distributed(
            verbosity,
            f"Epoch {epoch}: train {train_error.item():.6f}, "
            f"val {val_error.item():.6f}, test {test_error.item():.6f}")
        log(f"Epoch {epoch}: train {train_error.item():.6f}, "
            f"val {val_error.item():.6f}, test {test_error.item():.6f}")

        if checkpoint is not None and checkpoint(
                epoch, val_error.item()):
            save_model(model, optimizer, log_name)

        if early_stopping is not None:
            early_stopping(val_error.item())
            if early_stopping.early_stop:
                print_distributed(verbosity,
                                  f"Early stopping at epoch {epoch}")
                break
        total_epochs += 1
        if check_remaining_time(None, _time.time() - epoch_t0):
            print_distributed(verbosity, "Stopping: SLURM time limit near")
            break

    timer.stop()
    # final eval + plots (reference train_validate_test.py:470-520)
    if create_plots:
        err, tasks_err, tv, pv = test(test_loader, model, verbosity,
                                      precision)
        rank = dist.get_rank() if dist.is_initialized() else 0
        if rank == 0 and tv and tv[0].numel() > 0:
            from ..postprocess.visualizer import Visualizer
            viz = Visualizer(log_name,
                             num_heads=len(tv))
            viz.create_scatter_plots(tv, pv)
            viz.create_error_histograms(tv, pv)
    if os.getenv("HYDRAGNN_DUMP_TESTDATA"):
        err, tasks_err, tv, pv = test(test_loader, model, verbosity,
                                      precision)
        rank = dist.get_rank() if dist.is_initialized() else 0
        if rank == 0:
            import numpy as np
            os.makedirs(f"logs/{log_name}", exist_ok=True)
            np.savez(f"logs/{log_name}/testdata.npz",
                     **{f"true_{i}": t.cpu().numpy()
                        for i, t in enumerate(tv)},
                     **{f"pred_{i}": p_.cpu().numpy()
                        for i, p_ in enumerate(pv)})
    tr.save(f"logs/{log_name}")


# reference-named split entry points (train_validate_test.py:523-617)
def get_head_indices_graph(model, data):
    """Head index slices when every head is graph-level."""
    return get_head_indices(model, data)


def get_head_indices_node_or_mixed(model, data):
    """Head index slices with node-level or mixed heads."""
    return get_head_indices(model, data)


reduce_values_ranks_dist = reduce_values_ranks
reduce_values_ranks_mpi = reduce_values_ranks


# reference-named aliases (reference train_validate_test.py:609-655
# has separate torch-dist / MPI reduction entry points)
reduce_values_ranks_dist = reduce_values_ranks
reduce_values_ranks_mpi = reduce_values_ranks
