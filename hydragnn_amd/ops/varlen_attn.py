"""Segment-varlen attention for GPS global attention.

Dense-batch attention (reference hydragnn/globalAtt/gps.py:140) pads
every graph to the largest one in the batch; for molecular batches that
wastes both HBM traffic and FLOPs.  The HIP kernel
(csrc/varlen_attn.hip) runs one (graph, head) per workgroup with K/V
staged in LDS and an online softmax over exactly the graph's nodes.

Backward recomputes through a pure-torch reference (dense-batch
explicit-softmax attention with a finite mask bias — double
differentiable on every backend), so first and second order gradients
are exact without a hand-written backward kernel; the recompute only
triggers on the GPS training path, which is not the headline bench.
"""

from __future__ import annotations

import math

import torch

from ._extension import get_extension
from ..data import to_dense_batch

MAX_SEG = None  # r2: K/V tiling removed the segment cap
MAX_DH = 64


def torch_varlen_attention_chunked(q, k, v, batch, chunk=512):
    """Per-graph exact attention in query chunks: O(chunk * N_g)
    transient memory instead of the dense [B, maxN, maxN] logits —
    the recompute backward for LARGE segments (the dense reference
    would materialize maxN^2 per graph)."""
    N, H, dh = q.shape
    out = torch.empty_like(q)
    scale = 1.0 / math.sqrt(dh)
    n_graphs = int(batch.max()) + 1 if batch.numel() else 0
    counts = torch.bincount(batch, minlength=n_graphs)
    starts = torch.zeros(n_graphs, dtype=torch.long,
                         device=batch.device)
    if n_graphs > 1:
        starts[1:] = counts.cumsum(0)[:-1]
    for g in range(n_graphs):
        lo, n = int(starts[g]), int(counts[g])
        if n == 0:
            continue
        kg = k[lo:lo + n].transpose(0, 1)          # [H, n, dh]
        vg = v[lo:lo + n].transpose(0, 1)
        for c0 in range(0, n, chunk):
            qg = q[lo + c0:lo + min(c0 + chunk, n)].transpose(0, 1)
            logits = qg @ kg.transpose(-1, -2) * scale
            o = torch.softmax(logits, dim=-1) @ vg  # [H, c, dh]
            out[lo + c0:lo + c0 + o.shape[1]] = o.transpose(0, 1)
    return out


def torch_varlen_attention(q: torch.Tensor, k: torch.Tensor,
                           v: torch.Tensor, batch: torch.Tensor):
    """Reference path: [N, H, dh] q/k/v + per-node graph index ->
    [N, H, dh] via dense-batch masked SDPA."""
    N, H, dh = q.shape
    x = torch.cat([q, k, v], dim=1).reshape(N, 3 * H * dh)
    xd, mask = to_dense_batch(x, batch)
    B, Nmax = mask.shape
    qd, kd, vd = xd.view(B, Nmax, 3, H, dh).permute(2, 0, 3, 1, 4)
    # explicit softmax attention (not SDPA): double-differentiable on
    # every backend, which the recompute backward relies on; finite
    # mask bias keeps padded rows NaN-free
    bias = torch.where(mask.view(B, 1, 1, Nmax), 0.0, -1e9).to(q.dtype)
    logits = qd @ kd.transpose(-1, -2) / math.sqrt(dh) + bias
    out = torch.softmax(logits, dim=-1) @ vd
    out = out.permute(0, 2, 1, 3)  # [B, Nmax, H, dh]
    return out[mask]


class _VarlenAttn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, ptr, batch):
        ext = get_extension(required=True)
        out = ext.varlen_attention(q.contiguous(), k.contiguous(),
                                   v.contiguous(), ptr)
        ctx.save_for_backward(q, k, v, batch)
        return out

    @staticmethod
    def backward(ctx, g):
        # recompute through the torch reference, differentiating w.r.t.
        # the SAVED tensors (not detached copies) so the second-order
        # graph stays connected for force-style double backward
        q, k, v, batch = ctx.saved_tensors
        need = ctx.needs_input_grad[:3]
        max_seg = int(torch.bincount(batch).max()) \
            if batch.numel() else 0
        with torch.enable_grad():
            if max_seg > 1024:
                # dense recompute would materialize maxN^2 logits per
                # graph; chunked exact path keeps memory bounded
                out = torch_varlen_attention_chunked(q, k, v, batch)
            else:
                out = torch_varlen_attention(q, k, v, batch)
            inputs = [t for t, n in zip((q, k, v), need) if n]
            grads = iter(torch.autograd.grad(
                out, inputs, g, create_graph=torch.is_grad_enabled(),
                allow_unused=True))
        res = [next(grads) if n else None for n in need]
        return res[0], res[1], res[2], None, None


def varlen_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     ptr: torch.Tensor, batch: torch.Tensor):
    """[N, H, dh] q/k/v, graph rowptr [G+1] and per-node graph index
    [N] -> [N, H, dh].  HIP kernel on GPU (fp32), torch reference on
    CPU."""
    if not q.is_cuda:
        return torch_varlen_attention(q, k, v, batch)
    if q.dtype not in (torch.float32, torch.bfloat16):
        orig_dtype = q.dtype
        q, k, v = (t.float() for t in (q, k, v))
        return _VarlenAttn.apply(q, k, v, ptr, batch).to(orig_dtype)
    return _VarlenAttn.apply(q, k, v, ptr, batch)


def varlen_eligible(head_dim: int, max_seg: int, device) -> bool:
    import os
    if os.environ.get("HYDRAGNN_VARLEN_ATTN", "1") == "0":
        return False
    if not (isinstance(device, torch.device) and device.type == "cuda"):
        return False
    return head_dim <= MAX_DH
