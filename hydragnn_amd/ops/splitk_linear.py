"""Linear layer with a split-K weight gradient.

Per-edge MLPs (e.g. the MACE radial network) have tiny in/out widths
but E ~ 10^5-10^6 rows: the weight-gradient GEMM dW = X^T g has
M, N <= a few hundred and K = E, which hipBLASLt schedules on only
(M/16)x(N/16) workgroups — a serial-K crawl on a 256-CU chip.  Here the
K dimension is split into S chunks computed as a bmm (S x tiles
workgroups) and summed — an order of magnitude faster at these shapes.

backward() is built from differentiable torch ops, so the force
training double-backward works unchanged.
"""

from __future__ import annotations

import torch
from torch import nn


def _splitk_weight_grad(x2d: torch.Tensor, g2d: torch.Tensor,
                        chunks: int) -> torch.Tensor:
    E = x2d.shape[0]
    S = min(chunks, max(1, E // 256))
    if S <= 1 or not x2d.is_cuda:
        # split-K only pays on the GPU (hipBLASLt's serial-K underfill);
        # on CPU the chunked bmm is strictly slower than one mm
        return g2d.t() @ x2d
    pad = (S - E % S) % S
    if pad:
        x2d = torch.nn.functional.pad(x2d, (0, 0, 0, pad))
        g2d = torch.nn.functional.pad(g2d, (0, 0, 0, pad))
    xs = x2d.view(S, -1, x2d.shape[1])
    gs = g2d.view(S, -1, g2d.shape[1])
    return torch.bmm(gs.transpose(1, 2), xs).sum(0)


class _SplitKLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        return torch.nn.functional.linear(x, weight, bias)

    @staticmethod
    def backward(ctx, g):
        x, weight = ctx.saved_tensors
        g2d = g.reshape(-1, g.shape[-1])
        x2d = x.reshape(-1, x.shape[-1]).to(g2d.dtype)
        gx = gw = gb = None
        if ctx.needs_input_grad[0]:
            if g2d.is_cuda:
                # gx = g2d @ W == linear(g2d, W^T): route through the
                # split-K Function so the SECOND-order pass (force
                # double backward differentiating this product by W)
                # also gets a split-K weight grad — a plain mm here
                # regenerates the serial-K [out, E] @ [E, in] GEMM
                gx = _SplitKLinearFn.apply(
                    g2d, weight.t().to(g2d.dtype), None
                ).view_as(x).to(x.dtype)
            else:
                gx = (g2d @ weight.to(g2d.dtype)).view_as(x).to(x.dtype)
        if ctx.needs_input_grad[1]:
            gw = _splitk_weight_grad(x2d, g2d, 64).to(weight.dtype)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            gb = g2d.sum(0)
        return gx, gw, gb


class SplitKLinear(nn.Linear):
    def forward(self, x):
        if not x.is_cuda:
            # plain autograd linear on CPU (no custom Function
            # overhead; split-K is a GPU-shape fix)
            return torch.nn.functional.linear(x, self.weight,
                                              self.bias)
        return _SplitKLinearFn.apply(x, self.weight, self.bias)
