"""Optimizer selection incl. ZeRO-1 optimizer-state sharding
(reference: hydragnn/utils/optimizer/optimizer.py:53-123)."""

from __future__ import annotations

import torch
import torch.distributed as dist

_OPTS = {
    "SGD": torch.optim.SGD,
    "Adam": torch.optim.Adam,
    "AdamW": torch.optim.AdamW,
    "Adamax": torch.optim.Adamax,
    "RMSprop": torch.optim.RMSprop,
    "Adagrad": torch.optim.Adagrad,
    "Adadelta": torch.optim.Adadelta,
    "FusedAdam": torch.optim.AdamW,   # rocm fused path selected by torch
    "FusedLAMB": torch.optim.AdamW,
}


def select_optimizer(model, config):
    """config = config["NeuralNetwork"]["Training"]["Optimizer"]."""
    opt_type = config.get("type", "AdamW")
    lr = config.get("learning_rate", 1e-3)
    if opt_type not in _OPTS:
        raise ValueError(f"Unknown optimizer {opt_type}")
    cls = _OPTS[opt_type]
    kwargs = {"lr": lr}
    if opt_type in ("SGD",):
        kwargs["momentum"] = config.get("momentum", 0.9)
    use_zero = config.get("use_zero_redundancy", False)
    if use_zero and dist.is_initialized() and dist.get_world_size() > 1:
        from torch.distributed.optim import ZeroRedundancyOptimizer
        opt = ZeroRedundancyOptimizer(
            model.parameters(), optimizer_class=cls, **kwargs)
    else:
        opt = cls(model.parameters(), **kwargs)
    clip = config.get("grad_clip_norm")
    if clip is not None:
        # consumed by train() / the captured step before each
        # optimizer step (global-norm clipping)
        opt._hydragnn_grad_clip = float(clip)
    return opt


def select_standard_optimizer(model, config):
    """Non-sharded optimizer selection (reference optimizer.py
    split)."""
    cfg = dict(config)
    cfg["use_zero_redundancy"] = False
    return select_optimizer(model, cfg)


def select_zero_redundancy_optimizer(model, config):
    """ZeRO-1 sharded optimizer selection."""
    cfg = dict(config)
    cfg["use_zero_redundancy"] = True
    return select_optimizer(model, cfg)
