"""Rank-aware printing/logging (reference: hydragnn/utils/print/
print_utils.py:29-110 — verbosity-leveled print dispatch, rank-gated
tqdm, file+console logger under logs/<name>/run.log)."""

from __future__ import annotations

import logging
import os
import sys

import torch.distributed as dist

_logger = None


def _rank() -> int:
    if dist.is_initialized():
        return dist.get_rank()
    return int(os.getenv("RANK", "0"))


def print_master(*args, **kwargs):
    if _rank() == 0:
        print(*args, **kwargs)


def print_distributed(verbosity_level, *args, **kwargs):
    if int(verbosity_level) > 0 or _rank() == 0:
        if int(verbosity_level) >= 2 or _rank() == 0:
            print(f"[{_rank()}]", *args, **kwargs)


def iterate_tqdm(iterable, verbosity_level=0, **kwargs):
    if int(verbosity_level) >= 2 and _rank() == 0:
        try:
            from tqdm import tqdm
            return tqdm(iterable, **kwargs)
        except ImportError:
            pass
    return iterable


def setup_log(prefix: str, path: str = "./logs/"):
    global _logger
    d = os.path.join(path, prefix)
    os.makedirs(d, exist_ok=True)
    logger = logging.getLogger("hydragnn_amd")
    logger.setLevel(logging.INFO)
    logger.handlers.clear()
    fh = logging.FileHandler(os.path.join(d, "run.log"))
    fh.setFormatter(logging.Formatter(
        f"%(asctime)s [rank {_rank()}] %(message)s"))
    logger.addHandler(fh)
    if _rank() == 0:
        sh = logging.StreamHandler(sys.stdout)
        sh.setFormatter(logging.Formatter("%(message)s"))
        logger.addHandler(sh)
    _logger = logger
    return logger


def log(*args):
    if _logger is not None:
        _logger.info(" ".join(str(a) for a in args))


def log0(*args):
    if _rank() == 0:
        log(*args)


def print_nothing(*args, **kwargs):
    """Verbosity sink (reference print_utils)."""


def print_all_processes(*args, **kwargs):
    """Print from every rank, rank-prefixed."""
    import torch.distributed as dist
    rank = dist.get_rank() if dist.is_initialized() else 0
    print(f"[{rank}]", *args, **kwargs)
