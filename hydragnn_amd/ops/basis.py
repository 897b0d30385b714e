"""Radial basis functions and cutoff envelopes.

Covers the reference's radial zoo (SURVEY.md §2c "RBF/SBF bases"):
Bessel / Gaussian / Chebyshev / sinc bases and polynomial / cosine
cutoffs (/root/reference/hydragnn/utils/model/mace_utils/modules/
radial.py:33-154, models/SCFStack.py:67, models/PAINNStack.py:358).
Functional, differentiable (double-backward via composition); model
modules wrap these with trainable/buffered parameters.
"""

from __future__ import annotations

import math

import torch

__all__ = [
    "bessel_basis",
    "gaussian_basis",
    "chebyshev_basis",
    "sinc_basis",
    "polynomial_cutoff",
    "cosine_cutoff",
]


def bessel_basis(r: torch.Tensor, r_max: float,
                 weights: torch.Tensor) -> torch.Tensor:
    """e_n(r) = sqrt(2/r_max) * sin(w_n r) / r, w_n trainable
    (init n*pi/r_max).  r: [E, 1] -> [E, n_basis]."""
    prefactor = math.sqrt(2.0 / r_max)
    num = torch.sin(r * weights)  # [E, n]
    return prefactor * num / r.clamp(min=1e-10)


def gaussian_basis(r: torch.Tensor, offsets: torch.Tensor,
                   coeff: float) -> torch.Tensor:
    """exp(coeff * (r - mu)^2), coeff = -0.5/dr^2 (SchNet smearing)."""
    diff = r - offsets
    return torch.exp(coeff * diff * diff)


def chebyshev_basis(r: torch.Tensor, r_max: float,
                    num_basis: int) -> torch.Tensor:
    """Chebyshev polynomials of x = 2r/r_max - 1, T_1..T_num."""
    x = (2.0 * r / r_max - 1.0).clamp(-1.0, 1.0)
    outs = []
    t_prev = torch.ones_like(x)
    t_cur = x
    for _ in range(num_basis):
        outs.append(t_cur)
        t_next = 2.0 * x * t_cur - t_prev
        t_prev, t_cur = t_cur, t_next
    return torch.cat(outs, dim=-1)


def sinc_basis(r: torch.Tensor, r_cut: float, num_basis: int) -> torch.Tensor:
    """sin(n pi r / rc) / r for n=1..num (PaiNN sinc expansion).
    Safe at r->0 (limit n pi / rc)."""
    n = torch.arange(1, num_basis + 1, device=r.device, dtype=r.dtype)
    arg = n * math.pi * r / r_cut  # [E, n]
    small = r < 1e-8
    safe_r = torch.where(small, torch.ones_like(r), r)
    out = torch.sin(arg) / safe_r
    limit = n * math.pi / r_cut * torch.ones_like(r)
    return torch.where(small.expand_as(out), limit, out)


def polynomial_cutoff(r: torch.Tensor, r_max: float, p: float = 6.0
                      ) -> torch.Tensor:
    """MACE polynomial envelope: 1 - ((p+1)(p+2)/2) x^p + p(p+2) x^(p+1)
    - (p(p+1)/2) x^(p+2), zero beyond r_max."""
    x = r / r_max
    out = (
        1.0
        - ((p + 1.0) * (p + 2.0) / 2.0) * torch.pow(x, p)
        + p * (p + 2.0) * torch.pow(x, p + 1.0)
        - (p * (p + 1.0) / 2.0) * torch.pow(x, p + 2.0)
    )
    return out * (x < 1.0).to(r.dtype)


def cosine_cutoff(r: torch.Tensor, r_cut: float) -> torch.Tensor:
    """0.5 (cos(pi r / rc) + 1), zero beyond rc."""
    out = 0.5 * (torch.cos(math.pi * r / r_cut) + 1.0)
    return out * (r < r_cut).to(r.dtype)
