"""Minimal real-O(3) representation toolkit for the MACE stack.

Replaces the e3nn machinery the reference depends on
(/root/reference/hydragnn/utils/model/mace_utils/, irreps_tools.py)
with exactly what HydraGNN's MACE needs: uniform-multiplicity SH-like
irreps (Cx0e + Cx1o + Cx2e + ..., parity (-1)^l — what
create_irreps_string emits, irreps_tools.py:116), real Wigner-3j
coupling tensors, per-l block linear layers, and the uvu tensor-product
path table.

Conventions: real spherical harmonics as in ops/sph.py (Wikipedia real
SH; l=1 ordered (y,z,x)).  Wigner-3j tensors are computed once at
module import per (l1,l2,l3) from sympy's exact complex 3j symbols via
the numeric complex->real change of basis, cached, and verified by an
equivariance unit test (tests/test_mace_o3.py).
"""

from __future__ import annotations

import math
from functools import lru_cache
from typing import List, Tuple

import numpy as np
import torch
from torch import nn


def sh_slices(lmax: int) -> List[slice]:
    return [slice(l * l, (l + 1) * (l + 1)) for l in range(lmax + 1)]


def dim(lmax: int) -> int:
    return (lmax + 1) ** 2


# ---------------------------------------------------------------------------
# complex->real change of basis U_l:  Y^R_m = sum_m' U[m, m'] Y^C_{m'}
# rows ordered m = -l..l
# ---------------------------------------------------------------------------
@lru_cache(maxsize=None)
def _u_matrix(l: int) -> np.ndarray:
    U = np.zeros((2 * l + 1, 2 * l + 1), dtype=np.complex128)
    def idx(m):  # m -> row/col index
        return m + l
    s2 = 1.0 / math.sqrt(2.0)
    for m in range(-l, l + 1):
        if m == 0:
            U[idx(0), idx(0)] = 1.0
        elif m > 0:
            # sqrt2 (-1)^m Re Y^m = ((-1)^m Y^m + Y^-m)/sqrt2
            U[idx(m), idx(m)] = ((-1) ** m) * s2
            U[idx(m), idx(-m)] = s2
        else:
            mt = -m
            # sqrt2 (-1)^mt Im Y^mt = -i((-1)^mt Y^mt - Y^-mt)/sqrt2
            U[idx(m), idx(mt)] = -1j * ((-1) ** mt) * s2
            U[idx(m), idx(-mt)] = 1j * s2
    return U


@lru_cache(maxsize=None)
def _wigner_3j_np(l1: int, l2: int, l3: int) -> np.ndarray:
    """Real coupling tensor W[a, b, c] (a over l1, b over l2, c over l3)
    invariant under the real Wigner rotations of our SH basis."""
    from sympy.physics.wigner import wigner_3j

    T = np.zeros((2 * l1 + 1, 2 * l2 + 1, 2 * l3 + 1), dtype=np.complex128)
    for m1 in range(-l1, l1 + 1):
        for m2 in range(-l2, l2 + 1):
            m3 = -(m1 + m2)
            if abs(m3) > l3:
                continue
            T[m1 + l1, m2 + l2, m3 + l3] = float(
                wigner_3j(l1, l2, l3, m1, m2, m3))
    U1, U2, U3 = _u_matrix(l1), _u_matrix(l2), _u_matrix(l3)
    W = np.einsum("am,bn,co,mno->abc", U1, U2, U3, T)
    # result is real (even l1+l2+l3) or purely imaginary (odd): fix phase
    re, im = np.abs(W.real).max(), np.abs(W.imag).max()
    if im > re:
        W = W.imag
    else:
        W = W.real
    norm = np.sqrt((W ** 2).sum())
    if norm < 1e-12:
        return np.zeros_like(W, dtype=np.float64)
    return np.ascontiguousarray(W.astype(np.float64))


@lru_cache(maxsize=None)
def wigner_3j(l1: int, l2: int, l3: int) -> torch.Tensor:
    return torch.from_numpy(_wigner_3j_np(l1, l2, l3))


def allowed_l3(l1: int, l2: int, lmax: int,
               parity_sh: bool = True) -> List[int]:
    """Triangle-allowed outputs; with parity_sh, keep only l3 with
    parity (-1)^{l1+l2} (the SH-like irrep tower has parity (-1)^l)."""
    out = []
    for l3 in range(abs(l1 - l2), min(l1 + l2, lmax) + 1):
        if parity_sh and ((l1 + l2 + l3) % 2 != 0):
            continue
        out.append(l3)
    return out


class IrrepsLinear(nn.Module):
    """Per-l channel-mixing linear map on [N, C_in, D] -> [N, C_out, D]
    (the o3.Linear equivalent for uniform-multiplicity towers; bias only
    on l=0).  Weights are per-l [C_in, C_out] matrices -> a batched GEMM
    per l on MFMA via hipBLASLt."""

    def __init__(self, c_in: int, c_out: int, lmax: int,
                 bias: bool = False):
        super().__init__()
        self.lmax = lmax
        self.c_in = c_in
        self.c_out = c_out
        self.weight = nn.Parameter(
            torch.randn(lmax + 1, c_in, c_out) / math.sqrt(c_in))
        self.bias = nn.Parameter(torch.zeros(c_out)) if bias else None
        # per-m l index so the whole map is ONE bmm over D slices
        lmap = torch.cat([torch.full((2 * l + 1,), l, dtype=torch.long)
                          for l in range(lmax + 1)])
        self.register_buffer("lmap", lmap)

    def forward(self, x: torch.Tensor,
                add: "torch.Tensor | None" = None) -> torch.Tensor:
        from ...ops.irreps_linear import (irreps_linear,
                                          irreps_linear_eligible)
        if irreps_linear_eligible(x, self.weight):
            # single-pass MFMA kernel (csrc/irreps_linear.hip): x and
            # the whole weight stack LDS-resident, no permute copies;
            # optional residual fused into the epilogue
            return irreps_linear(x.to(torch.bfloat16).contiguous(),
                                 self.weight.to(torch.bfloat16),
                                 self.lmap, self.bias,
                                 add=None if add is None
                                 else add.to(torch.bfloat16))
        # fallback: one batched GEMM [D](N,C_in)@(C_in,C_out)
        W_m = self.weight.to(x.dtype)[self.lmap]  # [D, C_in, C_out]
        out = torch.bmm(x.permute(2, 0, 1), W_m).permute(1, 2, 0)
        if self.bias is not None:
            out = torch.cat([
                out[:, :, :1] + self.bias.to(x.dtype).view(1, -1, 1),
                out[:, :, 1:]], dim=-1)
        if add is not None:
            out = out + add.to(out.dtype)
        return out.contiguous()


def tp_paths(l_in_max: int, l_edge_max: int, l_out_max: int
             ) -> List[Tuple[int, int, int]]:
    """uvu instruction path table: (l1 from node feats, l2 from SH,
    l3 target), SH-parity filtered (irreps_tools.py:25 equivalent)."""
    paths = []
    for l1 in range(l_in_max + 1):
        for l2 in range(l_edge_max + 1):
            for l3 in allowed_l3(l1, l2, l_out_max):
                paths.append((l1, l2, l3))
    return paths


# ---------------------------------------------------------------------------
# Reference-named irreps tools (reference utils/model/irreps_tools.py
# and mace_utils/tools/cg.py) for this framework's dense
# uniform-multiplicity layout [N, C, (lmax+1)^2].
# ---------------------------------------------------------------------------
def create_irreps_string(num_channels: int, lmax: int) -> str:
    """e.g. 64x0e+64x1o+64x2e (alternating parity convention)."""
    return "+".join(f"{num_channels}x{l}{'e' if l % 2 == 0 else 'o'}"
                    for l in range(lmax + 1))


def tp_out_irreps_with_instructions(l_in_max: int, l_edge_max: int,
                                    l_out_max: int):
    """uvu path table plus per-path instruction tuples
    (l1_block, l2_block, path_index, 'uvu', True) — the reference's
    instruction shape over this framework's path list."""
    paths = tp_paths(l_in_max, l_edge_max, l_out_max)
    instructions = [(l1, l2, i, "uvu", True)
                    for i, (l1, l2, l3) in enumerate(paths)]
    return paths, instructions


def linear_out_irreps(lmax: int, lmax_target: int) -> int:
    """Output tower lmax after a linear map restricted to the
    target's l-content."""
    return min(lmax, lmax_target)


def reshape_irreps(x: torch.Tensor, num_channels: int,
                   lmax: int) -> torch.Tensor:
    """Flat e3nn-style [N, sum_l C*(2l+1)] -> dense [N, C, (lmax+1)^2]
    (the layout every block here consumes)."""
    n = x.shape[0]
    out = x.new_zeros(n, num_channels, (lmax + 1) ** 2)
    off = 0
    for l in range(lmax + 1):
        w = 2 * l + 1
        blk = x[:, off:off + num_channels * w].reshape(n, num_channels,
                                                       w)
        out[:, :, l * l:(l + 1) ** 2] = blk
        off += num_channels * w
    return out


def extract_invariant(x: torch.Tensor, num_layers: int = 1) -> torch.Tensor:
    """l=0 channel of a dense tower [N, C, D] -> [N, C] (reference
    extract_invariant on concatenated layer outputs)."""
    return x[:, :, 0]


def U_matrix_real(irreps_in, irreps_out, correlation: int,
                  **unused):
    """Reference cg.py entry-point name; the generalized
    Clebsch-Gordan contraction matrices for the symmetric
    contraction.  Accepts lmax ints for in/out on this framework's
    uniform-multiplicity towers."""
    from .symmetric_contraction import u_matrix
    lmax_in = int(irreps_in)
    lmax_out = int(irreps_out)
    return u_matrix(lmax_in, lmax_out, correlation)
