"""Coarse timers with distributed min/max/avg reductions
(reference: hydragnn/utils/profiling_and_tracing/time_utils.py:22-138)."""

from __future__ import annotations

import time

import torch
import torch.distributed as dist

_timers = {}


class Timer:
    def __init__(self, name: str):
        self.name = name
        self.elapsed = 0.0
        self._t0 = None
        _timers[name] = self

    def start(self):
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self._t0 = time.perf_counter()

    def stop(self):
        if self._t0 is None:
            return
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self.elapsed += time.perf_counter() - self._t0
        self._t0 = None

    def reduce(self):
        t = torch.tensor([self.elapsed])
        if dist.is_initialized() and dist.get_world_size() > 1:
            from ..distributed import to_comm_device
            t, _ = to_comm_device(t)
            tmin, tmax, tsum = t.clone(), t.clone(), t.clone()
            dist.all_reduce(tmin, op=dist.ReduceOp.MIN)
            dist.all_reduce(tmax, op=dist.ReduceOp.MAX)
            dist.all_reduce(tsum, op=dist.ReduceOp.SUM)
            return (float(tmin), float(tmax),
                    float(tsum) / dist.get_world_size())
        return float(t), float(t), float(t)


def print_timers(verbosity=0):
    rank = dist.get_rank() if dist.is_initialized() else 0
    for name, timer in _timers.items():
        tmin, tmax, tavg = timer.reduce()
        if rank == 0:
            print(f"Timer {name}: min {tmin:.4f}s max {tmax:.4f}s "
                  f"avg {tavg:.4f}s")


class TimerError(Exception):
    """Raised on Timer misuse (start twice / stop before start)."""
