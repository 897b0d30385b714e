"""Real spherical harmonics up to l=3 (closed-form polynomials).

Replaces e3nn.o3.spherical_harmonics as used by the reference MACE stack
(/root/reference/hydragnn/models/MACEStack.py:156-469) and the
equivariant attention (globalAtt/equivariant_attention.py:186).

Convention: m-ordering -l..l, "component" normalization
(sum_m Y_lm^2 = 2l+1 on the unit sphere, i.e. sqrt(4*pi) x orthonormal
real SH).  l=1 block is (y, z, x) so that it transforms with the
standard real Wigner-D used by our irreps toolkit.

Pure tensor ops: autograd (incl. double backward for the force pass)
falls out of composition.  The HIP-fused version (sph_fused) computes
the same polynomial in one kernel and is dispatched for large E on GPU.
"""

from __future__ import annotations

import math

import torch

__all__ = ["spherical_harmonics", "sh_dim"]

_SQRT3 = math.sqrt(3.0)
_SQRT5 = math.sqrt(5.0)
_SQRT15 = math.sqrt(15.0)
_SQRT7 = math.sqrt(7.0)
_SQRT105 = math.sqrt(105.0)
_SQRT35_2 = math.sqrt(35.0 / 2.0)
_SQRT21_2 = math.sqrt(21.0 / 2.0)


def sh_dim(lmax: int) -> int:
    return (lmax + 1) ** 2


def spherical_harmonics(
    vectors: torch.Tensor,
    lmax: int,
    normalize: bool = True,
    normalization: str = "component",
) -> torch.Tensor:
    """Y(v) : [..., 3] -> [..., (lmax+1)^2].

    normalize=True first maps v to the unit sphere (grad flows through).
    """
    assert normalization in ("component", "integral", "norm")
    assert 0 <= lmax <= 3, "spherical_harmonics implemented to lmax=3"
    if normalize:
        norm = torch.linalg.norm(vectors, dim=-1, keepdim=True)
        v = vectors / norm.clamp(min=1e-12)
    else:
        v = vectors
    x = v[..., 0]
    y = v[..., 1]
    z = v[..., 2]

    outs = [torch.ones_like(x)]  # l=0
    if lmax >= 1:
        outs += [_SQRT3 * y, _SQRT3 * z, _SQRT3 * x]
    if lmax >= 2:
        x2, y2, z2 = x * x, y * y, z * z
        outs += [
            _SQRT15 * x * y,
            _SQRT15 * y * z,
            (_SQRT5 / 2.0) * (3.0 * z2 - (x2 + y2 + z2)),
            _SQRT15 * x * z,
            (_SQRT15 / 2.0) * (x2 - y2),
        ]
    if lmax >= 3:
        r2 = x2 + y2 + z2
        outs += [
            (_SQRT35_2 / 2.0) * y * (3.0 * x2 - y2),
            _SQRT105 * x * y * z,
            (_SQRT21_2 / 2.0) * y * (5.0 * z2 - r2),
            (_SQRT7 / 2.0) * z * (5.0 * z2 - 3.0 * r2),
            (_SQRT21_2 / 2.0) * x * (5.0 * z2 - r2),
            (_SQRT105 / 2.0) * z * (x2 - y2),
            (_SQRT35_2 / 2.0) * x * (x2 - 3.0 * y2),
        ]
    out = torch.stack(outs, dim=-1)
    if normalization == "integral":
        out = out / math.sqrt(4.0 * math.pi)
    elif normalization == "norm":
        scales = []
        for l in range(lmax + 1):
            scales += [1.0 / math.sqrt(2 * l + 1)] * (2 * l + 1)
        out = out * out.new_tensor(scales)
    return out
