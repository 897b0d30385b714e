"""Segment reductions (scatter/gather) — the hot aggregation of every
message-passing stack.

Reference call sites: torch_scatter.scatter in MACE
(/root/reference/hydragnn/utils/model/mace_utils/modules/blocks.py:405),
index_add_ in PaiNN/PNAEq, scatter_add_ in EGNN, graph pooling in
Base.py.  Here they all funnel through one API backed by hand-written
CDNA4 HIP kernels on GPU and differentiable torch primitives on CPU.

Autograd composition gives double-backward (needed for forces =
-dE/dpos with create_graph=True) for free:
  scatter_sum backward = gather;  gather backward = scatter_sum.
Both directions are themselves custom autograd Functions, so the second
autograd pass re-enters the same HIP kernels.
"""

from __future__ import annotations

from typing import Optional

import torch

from ._extension import get_extension, use_eager

__all__ = ["scatter", "gather", "segment_softmax", "degree"]


def _use_hip(t: torch.Tensor) -> bool:
    return t.is_cuda and not use_eager()


# ---------------------------------------------------------------------------
# gather: out[e] = src[index[e]]
# ---------------------------------------------------------------------------
class _Gather(torch.autograd.Function):
    @staticmethod
    def forward(ctx, src: torch.Tensor, index: torch.Tensor,
                dim_size: int, backward_csr=None):
        ctx.save_for_backward(index)
        ctx.dim_size = src.shape[0]
        ctx.backward_csr = backward_csr
        if _use_hip(src):
            ext = get_extension(required=True)
            return ext.gather_fwd(src.contiguous(), index)
        return src.index_select(0, index)

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        (index,) = ctx.saved_tensors
        grad_src = scatter(grad_out, index, ctx.dim_size, reduce="sum",
                           csr=ctx.backward_csr)
        return grad_src, None, None, None


def gather(src: torch.Tensor, index: torch.Tensor,
           backward_csr=None) -> torch.Tensor:
    """out[e] = src[index[e]] with double-backward support.

    backward_csr: optional (perm, rowptr) precomputed sort of `index`
    (perm = argsort(index), rowptr over src rows) — the backward
    scatter then runs the contention-free indexed CSR kernel instead
    of atomics (the index is typically UNsorted here: gathers go by
    edge source while batches are destination-sorted)."""
    return _Gather.apply(src, index, src.shape[0], backward_csr)


# ---------------------------------------------------------------------------
# scatter: out[i] = reduce_e{index[e]==i} src[e]
# ---------------------------------------------------------------------------
def _rowptr_from_sorted(index: torch.Tensor, dim_size: int):
    """CSR row pointer from a sorted index vector, with no
    device-to-host sync (hipGraph-capturable)."""
    counts = torch.zeros(dim_size, dtype=torch.long, device=index.device)
    counts.index_add_(0, index, torch.ones_like(index))
    rowptr = torch.zeros(dim_size + 1, dtype=torch.long,
                         device=index.device)
    torch.cumsum(counts, 0, out=rowptr[1:])
    return rowptr


class _ScatterSum(torch.autograd.Function):
    @staticmethod
    def forward(ctx, src, index, dim_size, sorted_index=False, csr=None):
        ctx.csr = None
        if _use_hip(src):
            import os
            ext = get_extension(required=True)
            if csr is not None and \
                    os.environ.get("HYDRAGNN_CSR_SCATTER", "1") == "1":
                # precomputed index-sort: out[r] = sum src[perm[e]] —
                # contention-free and deterministic for UNSORTED index
                perm, rowptr = csr
                ctx.csr = csr
                ctx.save_for_backward(index)
                return ext.segment_sum_csr(src.contiguous(), rowptr,
                                           perm)
            if sorted_index and hasattr(ext, "segment_sum_csr") and \
                    os.environ.get("HYDRAGNN_CSR_SCATTER", "1") == "1":
                rowptr = _rowptr_from_sorted(index, dim_size)
                # pass the rowptr down the chain: the double-backward
                # scatter (gather's backward) reuses it instead of
                # falling back to atomics
                ctx.csr = (None, rowptr)
                ctx.save_for_backward(index)
                return ext.segment_sum_csr(src.contiguous(), rowptr)
            if os.environ.get("HYDRAGNN_DETERMINISTIC", "0") == "1":
                # order-independent accumulation: sort once, CSR reduce
                # (atomicAdd float accumulation is order-dependent)
                perm = torch.argsort(index, stable=True)
                rowptr = _rowptr_from_sorted(index[perm], dim_size)
                ctx.csr = (perm, rowptr)
                ctx.save_for_backward(index)
                return ext.segment_sum_csr(
                    src.index_select(0, perm).contiguous(), rowptr)
            ctx.save_for_backward(index)
            return ext.scatter_sum_fwd(src.contiguous(), index, dim_size)
        ctx.save_for_backward(index)
        out = src.new_zeros((dim_size,) + src.shape[1:])
        out.index_add_(0, index, src)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        (index,) = ctx.saved_tensors
        return (gather(grad_out, index, backward_csr=ctx.csr),
                None, None, None, None)


class _ScatterMean(torch.autograd.Function):
    @staticmethod
    def forward(ctx, src, index, dim_size, sorted_index=False):
        if _use_hip(src):
            import os
            ext = get_extension(required=True)
            if sorted_index and \
                    os.environ.get("HYDRAGNN_CSR_SCATTER", "1") == "1":
                # contention-free: CSR sum + rowptr-derived counts
                rowptr = _rowptr_from_sorted(index, dim_size)
                count = (rowptr[1:] - rowptr[:-1]).to(src.dtype)
                s = ext.segment_sum_csr(src.contiguous(), rowptr)
                out = s / count.clamp(min=1).view(
                    -1, *([1] * (src.dim() - 1)))
                ctx.save_for_backward(index, count)
                return out
            out, count = ext.scatter_mean_fwd(src.contiguous(), index, dim_size)
        else:
            out = src.new_zeros((dim_size,) + src.shape[1:])
            out.index_add_(0, index, src)
            count = torch.bincount(index, minlength=dim_size).to(src.dtype)
            out = out / count.clamp(min=1).view(-1, *([1] * (src.dim() - 1)))
        ctx.save_for_backward(index, count)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        index, count = ctx.saved_tensors
        inv = 1.0 / count.clamp(min=1)
        g = grad_out * inv.view(-1, *([1] * (grad_out.dim() - 1))).to(grad_out.dtype)
        return gather(g, index), None, None, None


class _ScatterMax(torch.autograd.Function):
    """max (or min) segment reduce. Backward routes gradient to argmax
    edges only. Not double-backward-critical (force models use sum)."""

    @staticmethod
    def forward(ctx, src, index, dim_size, is_max: bool):
        if _use_hip(src):
            ext = get_extension(required=True)
            out, arg = ext.scatter_minmax_fwd(
                src.contiguous(), index, dim_size, is_max)
        else:
            fill = float("-inf") if is_max else float("inf")
            out = src.new_full((dim_size,) + src.shape[1:], fill)
            red = "amax" if is_max else "amin"
            out = out.scatter_reduce(
                0, index.view(-1, *([1] * (src.dim() - 1))).expand_as(src),
                src, red, include_self=True)
            present = torch.bincount(index, minlength=dim_size) > 0
            out = torch.where(
                present.view(-1, *([1] * (src.dim() - 1))),
                out, torch.zeros_like(out))
            # argmax for backward
            gathered = out.index_select(0, index)
            is_arg = (gathered == src)
            eidx = torch.arange(src.shape[0], device=src.device)
            arg = torch.full(
                (dim_size,) + src.shape[1:], -1, dtype=torch.long,
                device=src.device)
            # first matching edge wins (reverse order so lowest index wins)
            flip = torch.flip(eidx, [0])
            src_f = src.flip(0)
            idx_f = index.flip(0)
            ga = out.index_select(0, idx_f)
            m = ga == src_f
            arg_src = torch.where(
                m, flip.view(-1, *([1] * (src.dim() - 1))).expand_as(src),
                torch.full_like(src, -1, dtype=torch.long)
                if src.dtype == torch.long else
                torch.full(src.shape, -1, dtype=torch.long, device=src.device))
            arg = arg.scatter_reduce(
                0, idx_f.view(-1, *([1] * (src.dim() - 1))).expand_as(arg_src),
                arg_src, "amax", include_self=True)
        ctx.save_for_backward(arg)
        ctx.src_shape = src.shape
        return out

    @staticmethod
    def backward(ctx, grad_out):
        (arg,) = ctx.saved_tensors
        grad_src = grad_out.new_zeros(ctx.src_shape)
        valid = arg >= 0
        flat_arg = arg.clamp(min=0)
        grad_src.scatter_add_(
            0, flat_arg, torch.where(valid, grad_out,
                                     torch.zeros_like(grad_out)))
        return grad_src, None, None, None


def scatter(
    src: torch.Tensor,
    index: torch.Tensor,
    dim_size: Optional[int] = None,
    reduce: str = "sum",
    dim: int = 0,
    sorted_index: bool = False,
    csr=None,
) -> torch.Tensor:
    """Segment reduction along dim 0.

    src   [E, ...]  values
    index [E]       destination ids in [0, dim_size)
    csr             optional (perm, rowptr) precomputed sort of index
                    (sum only): runs the indexed CSR kernel instead of
                    atomics even for unsorted index
    """
    assert dim == 0, "hydragnn_amd.ops.scatter reduces along dim 0"
    if dim_size is None:
        dim_size = int(index.max()) + 1 if index.numel() > 0 else 0
    if index.dtype != torch.long:
        index = index.long()
    if src.dim() == 1:
        out = scatter(src.unsqueeze(1), index, dim_size, reduce,
                      sorted_index=sorted_index, csr=csr)
        return out.squeeze(1)
    if reduce in ("sum", "add"):
        return _ScatterSum.apply(src, index, dim_size, sorted_index, csr)
    if reduce == "mean":
        return _ScatterMean.apply(src, index, dim_size, sorted_index)
    if reduce in ("max", "amax"):
        return _ScatterMax.apply(src, index, dim_size, True)
    if reduce in ("min", "amin"):
        return _ScatterMax.apply(src, index, dim_size, False)
    if reduce == "std":
        mean = scatter(src, index, dim_size, "mean")
        mean_sq = scatter(src * src, index, dim_size, "mean")
        var = (mean_sq - mean * mean).clamp(min=0)
        return torch.sqrt(var + 1e-5)
    raise ValueError(f"unknown reduce '{reduce}'")


def degree(index: torch.Tensor, dim_size: int,
           dtype: torch.dtype = torch.float32) -> torch.Tensor:
    return torch.bincount(index, minlength=dim_size).to(dtype)


def segment_softmax(src: torch.Tensor, index: torch.Tensor,
                    dim_size: Optional[int] = None) -> torch.Tensor:
    """Numerically-stable softmax over segments (per-target softmax used
    by the equivariant all-to-all attention,
    /root/reference/hydragnn/globalAtt/equivariant_attention.py:93)."""
    if dim_size is None:
        dim_size = int(index.max()) + 1 if index.numel() > 0 else 0
    seg_max = scatter(src.detach(), index, dim_size, "max")
    ex = torch.exp(src - gather(seg_max, index))
    denom = scatter(ex, index, dim_size, "sum")
    return ex / gather(denom, index).clamp(min=1e-16)
