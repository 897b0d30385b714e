"""MFMA-backed linear/matmul for per-edge MLP layers.

Dispatches bf16 GEMMs of the radial-MLP family (M = edges, N/K in
[64, 512]) to the hand-written MFMA kernel (csrc/mfma_linear.hip);
gradients are the same kernel with swapped operands plus the split-K
weight gradient, so force training's double backward stays on the
custom path.
"""

from __future__ import annotations

import torch

from ._extension import get_extension, use_eager
from .splitk_linear import SplitKLinear, _splitk_weight_grad




def _bf16_ok(x):
    """bf16 kernel eligibility: tensor already bf16, or autocast is on
    WITH bf16 as the autocast dtype (an fp16 session must not be
    silently rerouted through bf16 kernels)."""
    if x.dtype == torch.bfloat16:
        return True
    if not torch.is_autocast_enabled():
        return False
    try:
        return torch.get_autocast_dtype("cuda") == torch.bfloat16
    except (AttributeError, TypeError):
        return torch.get_autocast_gpu_dtype() == torch.bfloat16


def _eligible(M: int, N: int, K: int) -> bool:
    # forward C[M,N] = A[M,K] @ B^T needs K%32, N%64; the gA backward
    # flips roles (K'=N, N'=K) so both must satisfy both constraints
    return (N % 64 == 0 and K % 64 == 0 and N >= 64 and K >= 64
            and M >= 256)


class _MFMAMatmul(torch.autograd.Function):
    """C = A @ B (trans_b=False) or A @ B^T (trans_b=True), bf16."""

    @staticmethod
    def forward(ctx, A, B, trans_b):
        ctx.save_for_backward(A, B)
        ctx.trans_b = trans_b
        ext = get_extension(required=True)
        return ext.mfma_linear(A.contiguous(), B.contiguous(), None,
                               trans_b)

    @staticmethod
    def backward(ctx, g):
        A, B = ctx.saved_tensors
        g = g.contiguous()
        gA = gB = None
        if ctx.trans_b:
            # C = A @ B^T: gA = g @ B ; gB = g^T @ A
            if ctx.needs_input_grad[0]:
                gA = _MFMAMatmul.apply(g, B, False)
            if ctx.needs_input_grad[1]:
                gB = _splitk_weight_grad(A.reshape(-1, A.shape[-1]),
                                         g.reshape(-1, g.shape[-1]), 64)
        else:
            # C = A @ B: gA = g @ B^T ; gB = A^T @ g
            if ctx.needs_input_grad[0]:
                gA = _MFMAMatmul.apply(g, B, True)
            if ctx.needs_input_grad[1]:
                # _splitk_weight_grad(x2d, g2d) returns
                # [g2d_cols, x2d_cols]: (g, A) -> [K, N] = A^T g
                # directly (r2 fix: the old .t() silently transposed
                # the second-order weight grad; square shapes passed
                # the shape check with WRONG values)
                gB = _splitk_weight_grad(g.reshape(-1, g.shape[-1]),
                                         A.reshape(-1, A.shape[-1]),
                                         64).contiguous()
        return gA, gB, None


class _MFMALinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        xb = x.to(torch.bfloat16).contiguous()
        wb = weight.to(torch.bfloat16).contiguous()
        ctx.save_for_backward(xb, wb)
        ctx.has_bias = bias is not None
        ctx.w_dtype = weight.dtype
        ctx.x_dtype = x.dtype
        ext = get_extension(required=True)
        return ext.mfma_linear(xb, wb,
                               bias if bias is not None else None, True)

    @staticmethod
    def backward(ctx, g):
        xb, wb = ctx.saved_tensors
        g = g.contiguous().to(torch.bfloat16)
        gx = gw = gb = None
        if ctx.needs_input_grad[0]:
            gx = _MFMAMatmul.apply(g, wb, False).to(ctx.x_dtype)
        if ctx.needs_input_grad[1]:
            gw = _splitk_weight_grad(xb, g, 64).to(ctx.w_dtype)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            gb = g.float().sum(0)
        return gx, gw, gb


class _GemvLinearFn(torch.autograd.Function):
    """Linear with out_features <= 8: forward on the gemv_small_n
    kernel (hipBLASLt runs N=1 on MT1x4x256 tiles ~30x off roofline);
    backward composed of differentiable ops so the force-training
    double backward works unchanged."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        xb = x.to(torch.bfloat16).contiguous()
        wb = weight.to(torch.bfloat16).contiguous()
        ctx.save_for_backward(xb, wb)
        ctx.has_bias = bias is not None
        ctx.w_dtype = weight.dtype
        ctx.x_dtype = x.dtype
        ext = get_extension(required=True)
        return ext.gemv_small_n(xb, wb,
                                bias if bias is not None else None)

    @staticmethod
    def backward(ctx, g):
        xb, wb = ctx.saved_tensors
        g = g.contiguous().to(torch.bfloat16)
        gx = gw = gb = None
        if ctx.needs_input_grad[0]:
            if wb.shape[0] == 1:
                gx = (g * wb.reshape(-1)).to(ctx.x_dtype)
            else:
                gx = (g @ wb).to(ctx.x_dtype)
        if ctx.needs_input_grad[1]:
            if wb.shape[0] == 1:
                # dW[0,k] = sum_m g[m,0] x[m,k]: a weighted column sum
                # (the split-K bmm here is a degenerate M=1 batched
                # GEMM that hipBLASLt runs on MT1x4 tiles ~50x off
                # roofline)
                gw = (xb.float() * g.float()).sum(0, keepdim=True) \
                    .to(ctx.w_dtype)
            else:
                gw = _splitk_weight_grad(xb, g, 64).to(ctx.w_dtype)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            gb = g.float().sum(0)
        return gx, gw, gb


class MFMALinear(SplitKLinear):
    """nn.Linear drop-in: MFMA kernel when shapes/dtypes qualify,
    narrow-output GEMV kernel for out_features <= 8, split-K Linear
    otherwise."""

    def forward(self, x):
        hip_ok = (
            x.is_cuda and not use_eager()
            and _bf16_ok(x)
            and x.dim() == 2)
        if hip_ok and _eligible(x.shape[0], self.out_features,
                                self.in_features):
            return _MFMALinearFn.apply(x, self.weight, self.bias)
        if (hip_ok and self.out_features <= 8
                and self.in_features % 8 == 0 and x.shape[0] >= 256):
            return _GemvLinearFn.apply(x, self.weight, self.bias)
        return super().forward(x)
