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
transposed (trans_w flipped); gW is a small per-l reduction GEMM done
with torch einsum (output is only [Cin, Cout] per l).  First and
second order both stay functional, so force training works.
"""

from __future__ import annotations

import torch

from ._extension import get_extension, use_eager

_MAX_LDS = 160 * 1024


def irreps_kernel_ok(n: int, c_in: int, c_out: int, d: int,
                     n_l: int) -> bool:
    if c_in % 32 != 0 or c_out % 64 != 0 or d > 16:
        return False
    lds = (d * 16 * (c_in + 8) + n_l * c_in * (c_out + 8)) * 2
    return lds <= _MAX_LDS and n >= 64


class _IrrepsLinearFn(torch.autograd.Function):
    """out[n,o,m] = sum_i x[n,i,m] * Wk[lmap[m]][i,o], with
    Wk[i,o] = W[l,i,o] (trans_w=False) or W[l,o,i] (trans_w=True)."""

    @staticmethod
    def forward(ctx, x, W, lmap, bias, trans_w):
        ext = get_extension(required=True)
        out = ext.irreps_linear(x.contiguous(), W.contiguous(), lmap,
                                bias, trans_w)
        ctx.save_for_backward(x, W, lmap, bias)
        ctx.trans_w = trans_w
        return out

    @staticmethod
    def backward(ctx, g):
        x, W, lmap, bias = ctx.saved_tensors
        trans_w = ctx.trans_w
        g = g.contiguous()
        gx = gw = gb = None
        if ctx.needs_input_grad[0]:
            gx = _IrrepsLinearFn.apply(g, W, lmap, None, not trans_w)
        if ctx.needs_input_grad[1]:
            # lmap is the canonical l-block layout (lo = l^2), so the
            # slices are static — no device sync, capture-safe
            parts = []
            d = x.shape[2]
            for l in range(W.shape[0]):
                lo, hi = l * l, min((l + 1) * (l + 1), d)
                gw_l = torch.einsum("nim,nom->io", x[:, :, lo:hi],
                                    g[:, :, lo:hi])
                parts.append(gw_l.t() if trans_w else gw_l)
            gw = torch.stack(parts).to(W.dtype)
        if bias is not None and ctx.needs_input_grad[3]:
            gb = g[:, :, 0].sum(0).to(bias.dtype)
        return gx, gw, None, gb, None


def irreps_linear(x: torch.Tensor, W: torch.Tensor, lmap: torch.Tensor,
                  bias=None) -> torch.Tensor:
    """[N, Cin, D] x, [L, Cin, Cout] W -> [N, Cout, D] on the MFMA
    kernel (bf16)."""
    return _IrrepsLinearFn.apply(x, W, lmap, bias, False)


def irreps_linear_eligible(x: torch.Tensor, W: torch.Tensor) -> bool:
    import os
    if use_eager() or not x.is_cuda:
        return False
    if os.environ.get("HYDRAGNN_IRREPS_MFMA", "1") == "0":
        return False
    if not (x.dtype == torch.bfloat16 or torch.is_autocast_enabled()):
        return False
    return irreps_kernel_ok(x.shape[0], x.shape[1], W.shape[2],
                            x.shape[2], W.shape[0])
