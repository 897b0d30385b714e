"""Region tracer with pluggable backends.

Redesign of /root/reference/hydragnn/utils/profiling_and_tracing/
tracer.py:29-489: module-level start/stop/enable/disable/save with a
wall-clock timer backend (per-call history) and a ROCm-SMI energy
backend (reads GPU energy counters via rocm_smi, replacing the
reference's NVML/rocm/hwmon trio with the MI355X-native path).  Optional
GPU sync and barrier at region edges under HYDRAGNN_TRACE_LEVEL=1.
"""

from __future__ import annotations

import os
import time
from collections import defaultdict
from contextlib import contextmanager
from functools import wraps
from typing import Dict, List

import torch
import torch.distributed as dist

_enabled = False
_backends: List["TracerBackend"] = []


class TracerBackend:
    def start(self, name: str) -> None: ...
    def stop(self, name: str) -> None: ...
    def save(self, path: str, rank: int) -> None: ...


class WallTimer(TracerBackend):
    """GPTL-style wall-clock region timer with per-call history."""

    def __init__(self):
        self.t0: Dict[str, float] = {}
        self.history = defaultdict(list)

    def start(self, name):
        self.t0[name] = time.perf_counter()

    def stop(self, name):
        if name in self.t0:
            self.history[name].append(time.perf_counter() - self.t0.pop(name))

    def save(self, path, rank):
        fname = os.path.join(path, f"gp_timing.p{rank}")
        os.makedirs(path, exist_ok=True)
        with open(fname, "w") as f:
            f.write(f"{'region':<24}{'count':>8}{'total_s':>14}{'avg_s':>14}"
                    f"{'max_s':>14}\n")
            for name, hist in sorted(self.history.items()):
                tot = sum(hist)
                f.write(f"{name:<24}{len(hist):>8}{tot:>14.6f}"
                        f"{tot / len(hist):>14.6f}{max(hist):>14.6f}\n")
        # per-call history (the reference's gp_full.p<rank>), capped
        with open(os.path.join(path, f"gp_full.p{rank}"), "w") as f:
            for name, hist in sorted(self.history.items()):
                for i, dt in enumerate(hist[-10000:]):
                    f.write(f"{name} {i} {dt:.6f}\n")


class RocmEnergyTracer(TracerBackend):
    """Per-region GPU energy from the ROCm-SMI energy counter
    (reference tracer.py:210-288 reads the same counters via
    rocm_smi python bindings; we shell rocm-smi lazily and degrade to
    no-op off-GPU)."""

    def __init__(self):
        self.available = torch.cuda.is_available()
        self.e0: Dict[str, float] = {}
        self.energy = defaultdict(float)
        self.counts = defaultdict(int)
        self._smi = None
        if self.available:
            try:
                from amdsmi import (amdsmi_init, amdsmi_get_processor_handles,
                                    amdsmi_get_energy_count)
                amdsmi_init()
                self._handles = amdsmi_get_processor_handles()
                self._get = amdsmi_get_energy_count
            except Exception:
                self.available = False

    def _read(self) -> float:
        try:
            dev = torch.cuda.current_device()
            info = self._get(self._handles[dev])
            return float(info["energy_accumulator"]) * float(
                info.get("counter_resolution", 15.3)) * 1e-6  # J
        except Exception:
            return 0.0

    def start(self, name):
        if self.available:
            self.e0[name] = self._read()

    def stop(self, name):
        if self.available and name in self.e0:
            self.energy[name] += self._read() - self.e0.pop(name)
            self.counts[name] += 1

    def save(self, path, rank):
        if not self.energy:
            return
        os.makedirs(path, exist_ok=True)
        with open(os.path.join(path, f"gp_energy.p{rank}"), "w") as f:
            for name in sorted(self.energy):
                f.write(f"{name} {self.counts[name]} "
                        f"{self.energy[name]:.3f} J\n")


class GPTLTracer(TracerBackend):
    """Adapter over gptl4py (the reference's GPTL backend,
    reference tracer.py:29-120).  Import-gated: present on reference
    clusters, absent in the MI355X image."""

    def __init__(self):
        import gptl4py as gp
        self.gp = gp
        gp.initialize()

    def start(self, name):
        self.gp.start(name)

    def stop(self, name):
        self.gp.stop(name)

    def save(self, path, rank):
        os.makedirs(path, exist_ok=True)
        self.gp.pr_file(os.path.join(path, f"gp_timing.gptl.p{rank}"))
        if rank == 0 and hasattr(self.gp, "pr_summary_file"):
            self.gp.pr_summary_file(
                os.path.join(path, "gp_timing.gptl.summary"))


class ScorePTracer(TracerBackend):
    """Adapter over scorep.user regions (reference tracer.py Score-P
    backend).  Import-gated."""

    def __init__(self):
        import scorep.user as su
        self.su = su

    def start(self, name):
        self.su.region_begin(name)

    def stop(self, name):
        self.su.region_end(name)

    def save(self, path, rank):
        pass  # Score-P writes its own experiment archive


def initialize(energy: bool = False, verbose: bool = False,
               extra_backends=None):
    """Build the backend list: wall timer (+ per-call history) always;
    ROCm energy on request; GPTL / Score-P adapters when named in
    `extra_backends` or HYDRAGNN_TRACER_BACKENDS="gptl,scorep" AND
    importable (silently skipped otherwise, matching the reference's
    optional-import behavior)."""
    global _backends
    _backends = [WallTimer()]
    if energy:
        _backends.append(RocmEnergyTracer())
    names = set(extra_backends or [])
    env = os.environ.get("HYDRAGNN_TRACER_BACKENDS", "")
    names |= {n.strip() for n in env.split(",") if n.strip()}
    for name, cls in (("gptl", GPTLTracer), ("scorep", ScorePTracer)):
        if name in names:
            try:
                _backends.append(cls())
            except ImportError:
                if verbose:
                    print(f"[tracer] backend '{name}' unavailable")


def enable():
    global _enabled
    if not _backends:
        initialize()
    _enabled = True


def disable():
    global _enabled
    _enabled = False


def has(name: str) -> bool:
    return _enabled


def _edge_sync():
    if os.getenv("HYDRAGNN_TRACE_LEVEL", "0") == "1":
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        if dist.is_initialized():
            dist.barrier()


def start(name: str):
    if _enabled:
        _edge_sync()
        for b in _backends:
            b.start(name)


def stop(name: str):
    if _enabled:
        _edge_sync()
        for b in _backends:
            b.stop(name)


def save(path: str):
    if not _enabled:
        return
    rank = dist.get_rank() if dist.is_initialized() else 0
    for b in _backends:
        b.save(path, rank)


def profile(name: str):
    """@tr.profile decorator."""
    def deco(fn):
        @wraps(fn)
        def wrapper(*args, **kwargs):
            start(name)
            try:
                return fn(*args, **kwargs)
            finally:
                stop(name)
        return wrapper
    return deco


def reset():
    global _backends, _enabled
    _backends = []
    _enabled = False


@contextmanager
def timer(name: str):
    """Context manager sugar over start/stop (reference
    tracer.py:485)."""
    start(name)
    try:
        yield
    finally:
        stop(name)


# gptl4py-compat shims (reference gptl4py_dummy.py): the region tracer
# here fills the role of GPTL; these names keep drop-in scripts alive.
# (initialize() above already covers gptl's initialize.)
def finalize():
    disable()


def pr_file(path: str):
    save(os.path.dirname(path) or ".")


def pr_summary_file(path: str):
    save(os.path.dirname(path) or ".")


@contextmanager
def nvtx_timer(name: str):
    with timer(name):
        yield


# reference-named alias: the reference's abstract tracer base class is
# `Tracer` (tracer.py:29)
Tracer = TracerBackend
