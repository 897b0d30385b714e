"""torch.profiler wrapper gated by config Profile:{enable,target_epoch}
(reference: hydragnn/utils/profiling_and_tracing/profile.py:19-80).
Emits tensorboard traces; a no-op otherwise. Works under ROCm — the
"cuda" activity maps to HIP kernels."""

from __future__ import annotations

import os

import torch


class Profiler:
    def __init__(self, config=None):
        config = config or {}
        self.enabled = bool(config.get("enable", 0))
        self.target_epoch = int(config.get("target_epoch", 0))
        self.trace_dir = config.get("trace_dir", "./logs/profile_traces")
        self._prof = None
        self._active = False

    def set_epoch(self, epoch: int):
        if self._prof is not None:
            self._prof.stop()
            self._prof = None
            self._active = False
        if self.enabled and epoch == self.target_epoch:
            os.makedirs(self.trace_dir, exist_ok=True)
            activities = [torch.profiler.ProfilerActivity.CPU]
            if torch.cuda.is_available():
                activities.append(torch.profiler.ProfilerActivity.CUDA)
            self._prof = torch.profiler.profile(
                activities=activities,
                schedule=torch.profiler.schedule(wait=5, warmup=3, active=3),
                on_trace_ready=torch.profiler.tensorboard_trace_handler(
                    self.trace_dir),
            )
            self._prof.start()
            self._active = True

    def step(self):
        if self._prof is not None:
            self._prof.step()

    def stop(self):
        if self._prof is not None:
            self._prof.stop()
            self._prof = None
            self._active = False
