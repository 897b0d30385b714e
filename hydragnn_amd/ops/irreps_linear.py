"""MFMA-backed per-l channel mixing for irreps towers.

`IrrepsLinear` (models/mace/o3.py) is the o3.Linear equivalent on the
dense [N, C, D] layout and the biggest GEMM family after the tensor
products.  The torch route (bmm over D slices) pays 3x HBM traffic
(permute copy, D skinny hipBLASLt GEMMs at 1.4-3% MFMA issue density,
permute-back copy — see profiles/README.md PMC audit).  The HIP kernel
(csrc/irreps_linear.hip) does the whole map in one pass: x tile and the
full weight stack LDS-resident, all D GEMMs on
v_mfma_f32_16x16x32_bf16, registers holding every m until the single
output write.

Gradient closure: gX is the same kernel with the weight read
transposed (trans_w flipped); gW is a second HIP kernel
(irreps_linear_gw_kernel) contracting over n with MFMA from the same
LDS staging — copy-free and deterministic (per-block fp32 partials,
summed on the stream).  An earlier bmm/einsum gW was measured SLOWER
than the torch path end-to-end (permuted copies of x and g); the
kernel version flips the A/B to +1.3% (32.7k vs 32.3k g/s, default
bench).  First and second order both stay on the kernels, so force
training works.
"""

from __future__ import annotations

import torch

from ._extension import get_extension, use_eager

_MAX_LDS = 160 * 1024


def irreps_kernel_ok(n: int, c_in: int, c_out: int, d: int,
                     n_l: int) -> bool:
    if c_in % 32 != 0 or c_out % 64 != 0 or d > 16:
        return False
    lds = (d * 32 * (c_in + 8) + n_l * c_in * (c_out + 8)) * 2
    return lds <= _MAX_LDS and n >= 64


class _IrrepsLinearFn(torch.autograd.Function):
    """out[n,o,m] = sum_i x[n,i,m] * Wk[lmap[m]][i,o], with
    Wk[i,o] = W[l,i,o] (trans_w=False) or W[l,o,i] (trans_w=True)."""

    @staticmethod
    def forward(ctx, x, W, lmap, bias, trans_w, add=None):
        ext = get_extension(required=True)
        out = ext.irreps_linear(x.contiguous(), W.contiguous(), lmap,
                                bias, trans_w, add)
        ctx.save_for_backward(x, W, lmap, bias)
        ctx.trans_w = trans_w
        return out

    @staticmethod
    def backward(ctx, g):
        x, W, lmap, bias = ctx.saved_tensors
        trans_w = ctx.trans_w
        g = g.contiguous()
        gx = gw = gb = None
        ga = g if (len(ctx.needs_input_grad) > 5
                   and ctx.needs_input_grad[5]) else None
        if ctx.needs_input_grad[0]:
            gx = _IrrepsLinearFn.apply(g, W, lmap, None, not trans_w)
        if ctx.needs_input_grad[1]:
            cin, cout = x.shape[1], g.shape[1]
            tiles = (cin // 16) * (cout // 16)
            if (cin % 16 == 0 and cout % 16 == 0 and tiles % 4 == 0
                    and tiles <= 16 and W.shape[0] <= 4):
                # copy-free MFMA contraction over n: per-block fp32
                # partials (deterministic), summed here
                ext = get_extension(required=True)
                nblocks = min(512, (x.shape[0] + 31) // 32)
                parts = ext.irreps_linear_gw(x.contiguous(), g, lmap,
                                             W.shape[0], nblocks)
                gw = parts.sum(0)
            else:
                # fallback: per-m batched GEMM + fold m -> l
                gw_m = torch.bmm(x.permute(2, 1, 0), g.permute(2, 0, 1))
                gw = gw_m.new_zeros(W.shape[0], cin, cout)
                gw.index_add_(0, lmap, gw_m)
            if trans_w:
                gw = gw.transpose(1, 2)
            gw = gw.to(W.dtype)
        if bias is not None and ctx.needs_input_grad[3]:
            gb = g[:, :, 0].sum(0).to(bias.dtype)
        return gx, gw, None, gb, None, ga


def irreps_linear(x: torch.Tensor, W: torch.Tensor, lmap: torch.Tensor,
                  bias=None, add=None) -> torch.Tensor:
    """[N, Cin, D] x, [L, Cin, Cout] W -> [N, Cout, D] on the MFMA
    kernel (bf16); `add` is an optional residual fused into the
    epilogue (its gradient is the identity)."""
    return _IrrepsLinearFn.apply(x, W, lmap, bias, False, add)


def irreps_linear_eligible(x: torch.Tensor, W: torch.Tensor) -> bool:
    import os
    if use_eager() or not x.is_cuda:
        return False
    if os.environ.get("HYDRAGNN_IRREPS_MFMA", "1") == "0":
        return False
    from .mfma_linear import _bf16_ok
    if not _bf16_ok(x):
        return False
    return irreps_kernel_ok(x.shape[0], x.shape[1], W.shape[2],
                            x.shape[2], W.shape[0])
