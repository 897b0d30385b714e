"""hipGraph-captured training step.

The MACE/MLIP training step at MD17 molecule sizes is launch-bound:
thousands of small kernels per fwd + double-backward.  On MI355X we
capture the whole fwd + backward (including the force double-backward)
in ONE hipGraph and replay it per batch, copying each new batch into
static device buffers first.  This is the MI355X-native replacement
for the reference's eager per-batch loop
(/root/reference/hydragnn/train/train_validate_test.py:658-835) — HIP
graphs instead of a tracing compiler, as the hardware model intends.

Requirements:
- static batch shapes (use preprocess/static_batch.StaticShapeCollater);
- CUDA/HIP device; no GradScaler (bf16/fp32/fp64 paths);
- the optimizer step and the fp32 gradient all-reduce
  (parallel/grad_sync.FlatGradSync) stay OUTSIDE the graph: they are a
  handful of foreach/RCCL launches, and keeping collectives out of
  capture is deliberate (capture-safety over the last few launches).

DDP interop: if the model is torch DDP-wrapped, capture runs under
``no_sync()`` so the reducer hooks stay out of the graph; gradient
averaging is done by FlatGradSync in fp32 after each replay.
"""

from __future__ import annotations

import os
from contextlib import nullcontext
from typing import Callable

import torch

from ..parallel.grad_sync import FlatGradSync

# Bookkeeping keys never copied into static buffers.
_SKIP_COPY = {"edges_sorted_", "num_graphs_", "num_nodes_",
              "num_real_graphs_", "static_shape_"}


def capture_enabled(batch=None, model=None) -> bool:
    """Captured stepping is on for CUDA devices when the batch is
    static-shape collated AND the model declares its forward
    shape-static (``supports_hipgraph_capture`` — replaying a graph
    that rebuilds neighbor lists or triplets internally, e.g. SchNet /
    DimeNet / dense-batch attention, would silently freeze
    data-dependent shapes).  HYDRAGNN_CAPTURE=1 forces, =0 disables."""
    env = os.environ.get("HYDRAGNN_CAPTURE")
    if env == "0":
        return False
    if not torch.cuda.is_available():
        return False
    if env == "1":
        return True
    if model is not None:
        base = model.module if hasattr(model, "module") else model
        if not getattr(base, "supports_hipgraph_capture", False):
            return False
    return bool(batch is not None and batch.get("static_shape_"))


class CapturedTrainStep:
    """Capture fwd+bwd once; per batch: H2D copy-in -> replay ->
    grad sync -> optimizer step."""

    def __init__(self, model, opt, static_batch, autocast,
                 compute_loss: Callable, param_dtype,
                 warmup_iters: int = 3):
        self.model = model
        self.opt = opt
        self.static = static_batch
        self.param_dtype = param_dtype
        self._shapes = {
            k: tuple(v.shape) for k, v in static_batch.items()
            if torch.is_tensor(v)
        }

        no_sync = getattr(model, "no_sync", nullcontext)

        # Pre-assign every grad as a view into ONE flat buffer: the
        # per-replay zeroing is a single fill kernel instead of ~100,
        # and the DP sync / a fused optimizer can address the whole
        # gradient contiguously.  A FusedAdamW optimizer already owns
        # such a buffer (ops/fused_adamw.py) — reuse it.
        params = [p for p in model.parameters() if p.requires_grad]
        if getattr(opt, "flat_grad", None) is not None:
            self._flat_grad = opt.flat_grad
        else:
            total = sum(p.numel() for p in params)
            dev = next(iter(params)).device if params else "cuda"
            self._flat_grad = torch.zeros(total, dtype=param_dtype,
                                          device=dev)
            off = 0
            for p in params:
                p.grad = self._flat_grad[off:off + p.numel()].view_as(p)
                off += p.numel()

        def fwd_bwd():
            self._flat_grad.zero_()
            self.static.pos.requires_grad_(True)
            with autocast:
                loss, tasks_loss, _ = compute_loss(self.static)
            loss.backward()
            return loss, tasks_loss

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side), no_sync():
            for _ in range(warmup_iters):
                loss, _ = fwd_bwd()
        # Drop every reference to the warmup autograd graph BEFORE
        # capturing: a live warmup loss keeps the side-stream
        # AccumulateGrad nodes alive, the capture-time backward reuses
        # them, and the resulting cross-stream sync during capture
        # intermittently corrupts the captured gradients (observed as
        # a ~1e30 loss on the sanity replay -> eager fallback).
        del loss
        torch.cuda.current_stream().wait_stream(side)
        self._grads = [p.grad for p in params if p.grad is not None]

        self.graph = torch.cuda.CUDAGraph()
        with no_sync():
            # capture on the SAME stream the warmups ran on, so any
            # autograd node that does survive warmup has a matching
            # canonical stream
            with torch.cuda.graph(self.graph, stream=side):
                self.loss, self.tasks_loss = fwd_bwd()

        self.grad_sync = FlatGradSync(
            [p for p in model.parameters() if p.grad is not None])
        # sanity replay: finite loss required
        self.graph.replay()
        lv = float(self.loss.detach().float().cpu())
        if not (lv == lv and abs(lv) < 1e30):
            raise RuntimeError(f"captured step produced loss {lv}")

    def matches(self, data) -> bool:
        for k, shape in self._shapes.items():
            v = data.get(k)
            if not torch.is_tensor(v) or tuple(v.shape) != shape:
                return False
        return True

    @torch.no_grad()
    def _copy_in(self, data):
        for k, dst in self.static.items():
            if k in _SKIP_COPY or not torch.is_tensor(dst):
                continue
            src = data.get(k)
            if src is dst:
                continue  # first batch: already the static storage
            if torch.is_floating_point(dst) and src.dtype != dst.dtype:
                src = src.to(dst.dtype)
            dst.copy_(src, non_blocking=True)

    def step(self, data):
        """One training step on ``data`` (a CPU pinned or device
        batch with shapes matching the captured one)."""
        self._copy_in(data)
        self.graph.replay()
        self.grad_sync()
        clip = getattr(self.opt, "_hydragnn_grad_clip", None)
        if clip is not None:
            # one global-norm clip over the flat gradient buffer
            norm = self._flat_grad.float().norm()
            scale = (clip / (norm + 1e-6)).clamp(max=1.0)
            self._flat_grad.mul_(scale.to(self._flat_grad.dtype))
        self.opt.step()
        return self.loss, self.tasks_loss


def get_or_build_stepper(model, opt, device_batch, autocast,
                         compute_loss, param_dtype):
    """Build (or fetch the cached) CapturedTrainStep for this
    model+optimizer.  Returns None if capture fails (caller falls back
    to eager). The cache lives on the model so warmup and timed epochs
    share one graph."""
    base = model.module if hasattr(model, "module") else model
    cached = getattr(base, "_hip_captured_step", None)
    if cached is not None:
        if cached is False:
            return None
        if cached.opt is opt and cached.matches(device_batch):
            return cached
    stepper = None
    err = None
    try:
        stepper = CapturedTrainStep(model, opt, device_batch, autocast,
                                    compute_loss, param_dtype)
    except Exception as e:  # pragma: no cover - GPU-only path
        err = e

    # All ranks must agree: a mix of captured (explicit FlatGradSync
    # all-reduce) and eager (DDP-hook all-reduce) ranks would mismatch
    # collectives and hang. Fall back everywhere unless all captured.
    import torch.distributed as dist
    if dist.is_initialized() and dist.get_world_size() > 1:
        ok = torch.tensor([0 if stepper is None else 1],
                          device=device_batch.pos.device
                          if torch.is_tensor(device_batch.get("pos"))
                          else None)
        dist.all_reduce(ok, op=dist.ReduceOp.MIN)
        if int(ok.item()) == 0:
            stepper = None

    if stepper is None:  # pragma: no cover - GPU-only path
        import sys
        print(f"[captured] hipGraph capture disabled "
              f"({err if err is not None else 'peer rank failed'}); "
              "eager fallback", file=sys.stderr)
        base._hip_captured_step = False
        return None
    base._hip_captured_step = stepper
    return stepper
