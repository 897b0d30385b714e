"""Checkpoint save/load, EarlyStopping, Checkpoint gating, tensorboard
writer.

On-disk format compatibility with the reference (SURVEY.md §5
"Checkpoint / resume"): single .pk file torch.save dict with
model_state_dict / optimizer_state_dict keys, per-epoch filenames +
latest symlink under logs/<name>/ (reference: hydragnn/utils/model/
model.py:106-313, 515-573)."""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist


def _rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def get_summary_writer(name: str, path: str = "./logs/"):
    if _rank() != 0:
        return None
    try:
        from torch.utils.tensorboard import SummaryWriter
        return SummaryWriter(os.path.join(path, name))
    except ImportError:
        return None


def _unwrap(model):
    return model.module if hasattr(model, "module") else model


def _full_state_dict(m):
    """FSDP1 full-state-dict gather when applicable (reference
    model.py:142)."""
    try:
        from torch.distributed.fsdp import (
            FullyShardedDataParallel as FSDP, StateDictType,
            FullStateDictConfig)
        if isinstance(m, FSDP):
            cfg = FullStateDictConfig(offload_to_cpu=True,
                                      rank0_only=True)
            with FSDP.state_dict_type(m, StateDictType.FULL_STATE_DICT,
                                      cfg):
                return m.state_dict()
    except ImportError:
        pass
    sd = m.state_dict()
    # FSDP2 DTensor shards -> full tensors
    out = {}
    for k, v in sd.items():
        if hasattr(v, "full_tensor"):
            v = v.full_tensor()
        out[k] = v
    return out


def save_model(model, optimizer, name: str, epoch: Optional[int] = None,
               path: str = "./logs/") -> None:
    """torch.save {model_state_dict, optimizer_state_dict} to
    logs/<name>/<name>[_epoch_E].pk + latest symlink; MultiTaskModelMP
    saves encoder plus a per-branch decoder file (branch-group rank0,
    reference model.py:66-189)."""
    d = os.path.join(path, name)
    os.makedirs(d, exist_ok=True)
    if hasattr(optimizer, "consolidate_state_dict"):
        optimizer.consolidate_state_dict()
    from ...models.multitask_mp import MultiTaskModelMP
    if isinstance(_unwrap(model), MultiTaskModelMP) or \
            isinstance(model, MultiTaskModelMP):
        mt = model if isinstance(model, MultiTaskModelMP) else \
            _unwrap(model)
        enc = mt.encoder.module if hasattr(mt.encoder, "module") \
            else mt.encoder
        dec = mt.decoder.module if hasattr(mt.decoder, "module") \
            else mt.decoder
        if _rank() == 0:
            torch.save({"model_state_dict": enc.state_dict(),
                        "optimizer_state_dict":
                        optimizer.state_dict()
                        if optimizer is not None else {}},
                       os.path.join(d, f"{name}.pk"))
        branch_rank = dist.get_rank(mt.branch_group) \
            if (dist.is_initialized() and mt.branch_group is not None) \
            else 0
        if branch_rank == 0:
            torch.save({"model_state_dict": dec.state_dict()},
                       os.path.join(
                           d, f"{name}_branch{mt.branch_id}.pk"))
        if dist.is_initialized():
            dist.barrier()
        return
    m = _unwrap(model)
    if _rank() == 0:
        fname = (f"{name}_epoch_{epoch}.pk" if epoch is not None
                 else f"{name}.pk")
        fpath = os.path.join(d, fname)
        torch.save({
            "model_state_dict": _full_state_dict(m),
            "optimizer_state_dict": optimizer.state_dict()
            if optimizer is not None else {},
        }, fpath)
        latest = os.path.join(d, f"{name}.pk")
        if epoch is not None and fpath != latest:
            if os.path.islink(latest) or os.path.exists(latest):
                os.remove(latest)
            os.symlink(fname, latest)
    if dist.is_initialized():
        dist.barrier()


def load_existing_model(model, name: str, path: str = "./logs/",
                        optimizer=None, map_location=None) -> None:
    fpath = os.path.join(path, name, f"{name}.pk")
    if map_location is None:
        map_location = "cpu"
    ckpt = torch.load(fpath, map_location=map_location, weights_only=False)
    state = ckpt["model_state_dict"]
    m = _unwrap(model)
    # module-prefix fixup (checkpoints saved from DDP-wrapped models)
    if any(k.startswith("module.") for k in state):
        state = {k.removeprefix("module."): v for k, v in state.items()}
    m.load_state_dict(state)
    if optimizer is not None and "optimizer_state_dict" in ckpt:
        optimizer.load_state_dict(ckpt["optimizer_state_dict"])


def load_existing_model_config(model, config, path: str = "./logs/",
                               optimizer=None) -> None:
    if config.get("continue", 0):
        name = config.get("startfrom")
        if name:
            load_existing_model(model, name, path, optimizer)


class EarlyStopping:
    """Stop when validation loss has not improved for `patience` epochs
    (reference model.py:515-530)."""

    def __init__(self, patience: int = 10, min_delta: float = 0.0):
        self.patience = patience
        self.min_delta = min_delta
        self.counter = 0
        self.best = None
        self.early_stop = False

    def __call__(self, val_loss: float) -> bool:
        if self.best is None or val_loss < self.best - self.min_delta:
            self.best = val_loss
            self.counter = 0
        else:
            self.counter += 1
            if self.counter >= self.patience:
                self.early_stop = True
        return self.early_stop


class Checkpoint:
    """Best-val-metric gated checkpointing with warmup
    (reference model.py:533-573)."""

    def __init__(self, name: str, warmup: int = 0, path: str = "./logs/"):
        self.name = name
        self.warmup = warmup
        self.path = path
        self.best = None

    def __call__(self, epoch: int, val_loss: float) -> bool:
        if epoch < self.warmup:
            return False
        if self.best is None or val_loss < self.best:
            self.best = val_loss
            return True
        return False


def print_model(model):
    """Layer-by-layer parameter sizes (reference model.py:453)."""
    from ..print.print_utils import print_master
    num_params = 0
    num_bytes = 0
    for k, v in _unwrap(model).state_dict().items():
        print_master("%50s\t%20s\t%10d" % (k, list(v.shape), v.numel()))
        num_params += v.numel()
        num_bytes += v.numel() * v.element_size()
    print_master("-" * 50)
    print_master("%50s\t%20s\t%10d" % ("Total", "", num_params))
    print_master("All (total, MB): %d %g"
                 % (num_params, num_bytes / 1024 / 1024))
    return num_params


def print_optimizer(optimizer):
    """Optimizer state summary (reference model.py pattern)."""
    from ..print.print_utils import print_master
    for i, group in enumerate(optimizer.param_groups):
        n = sum(p.numel() for p in group["params"])
        print_master(f"param_group {i}: {len(group['params'])} tensors,"
                     f" {n} params, lr={group.get('lr')}")


def tensor_divide(x1, x2):
    """Elementwise division with 0 where the denominator is 0."""
    out = torch.zeros_like(x1)
    mask = x2 != 0
    out[mask] = x1[mask] / x2[mask]
    return out


def unsorted_segment_mean(data, segment_ids, num_segments):
    """Segment mean by scatter (reference model.py:443); the HIP path
    is ops.scatter(..., 'mean')."""
    from ...ops import scatter
    return scatter(data, segment_ids, num_segments, "mean")


def calculate_PNA_degree(dataset, max_neighbours=None):
    """In-degree histogram for PNA scalers (dist-reduced when a
    process group is initialized)."""
    from ..config.config_utils import _gather_deg
    return _gather_deg(dataset)


def calculate_avg_deg(dataset):
    from ..config.config_utils import _calculate_avg_deg
    return _calculate_avg_deg(dataset)


calculate_PNA_degree_dist = calculate_PNA_degree
calculate_PNA_degree_mpi = calculate_PNA_degree
calculate_avg_deg_dist = calculate_avg_deg
calculate_avg_deg_mpi = calculate_avg_deg


def multitask_optim_state_dict(dual_optimizer):
    """Consolidated state dicts for the branch model-parallel
    DualOptimizer (reference model.py multitask pattern)."""
    out = {}
    for name in ("optimizer1", "optimizer2"):
        opt = getattr(dual_optimizer, name, None)
        if opt is not None and hasattr(opt, "state_dict"):
            out[name] = opt.state_dict()
    if not out and hasattr(dual_optimizer, "state_dict"):
        out = {"optimizer": dual_optimizer.state_dict()}
    return out


# reference-named aliases: the reference exposes separate torch-dist /
# MPI reduction variants (utils/model/model.py:357-440); ours selects
# the aggregation plane via HYDRAGNN_AGGR_BACKEND inside one function.
calculate_PNA_degree_dist = calculate_PNA_degree
calculate_PNA_degree_mpi = calculate_PNA_degree
calculate_avg_deg_dist = calculate_avg_deg
calculate_avg_deg_mpi = calculate_avg_deg
