"""Distributed linear regression of per-element reference energies.

Reference: hydragnn/preprocess/energy_linear_regression.py:28-208 —
least-squares (via normal equations, all-reduced across ranks, solved
with SVD) fit of E_total ~ sum_z n_z * e_z, used to shift energies
before MLIP training."""

from __future__ import annotations

from typing import Dict, Iterable, List, Optional

import numpy as np
import torch
import torch.distributed as dist


def composition_counts(z: torch.Tensor, num_elements: int = 119
                       ) -> np.ndarray:
    out = np.zeros(num_elements)
    vals, counts = torch.unique(z, return_counts=True)
    out[vals.numpy()] = counts.numpy()
    return out


def energy_linear_regression(dataset: Iterable,
                             num_elements: int = 119,
                             distributed: bool = True):
    """Fit per-element energies over a (possibly rank-sharded) dataset.
    Returns (e_per_element [num_elements], present_mask)."""
    AtA = np.zeros((num_elements, num_elements))
    Atb = np.zeros(num_elements)
    for d in dataset:
        z = d.get("z")
        if z is None:
            z = d.x[:, 0].long()
        energy = d.get("energy")
        if energy is None:
            energy = d.y
        row = composition_counts(z.flatten(), num_elements)
        AtA += np.outer(row, row)
        Atb += row * float(energy.flatten()[0])
    if distributed and dist.is_initialized() and \
            dist.get_world_size() > 1:
        from ..utils.distributed import to_comm_device
        t, _ = to_comm_device(torch.from_numpy(AtA))
        dist.all_reduce(t)
        AtA = t.cpu().numpy()
        t, _ = to_comm_device(torch.from_numpy(Atb))
        dist.all_reduce(t)
        Atb = t.cpu().numpy()
    present = np.diag(AtA) > 0
    e = np.zeros(num_elements)
    if present.any():
        sub = AtA[np.ix_(present, present)]
        e[present] = np.linalg.lstsq(sub, Atb[present], rcond=None)[0]
    return e, present


def shift_energies(dataset: Iterable, e_per_element: np.ndarray) -> None:
    """Subtract the fitted composition baseline in place."""
    for d in dataset:
        z = d.get("z")
        if z is None:
            z = d.x[:, 0].long()
        baseline = float(e_per_element[z.flatten().numpy()].sum())
        if d.get("energy") is not None:
            d.energy = d.energy - baseline
        if d.get("y") is not None and d.y.numel() == 1:
            d.y = d.y - baseline


def solve_least_squares_svd(A, b):
    """Pseudo-inverse least squares via SVD (reference
    energy_linear_regression.py:28); numpy in, numpy out."""
    import numpy as np
    U, S, Vt = np.linalg.svd(np.asarray(A, dtype=float),
                             full_matrices=False)
    keep = S > 1e-12 * (S[0] if S.size else 1.0)
    S_inv = np.zeros_like(S)
    S_inv[keep] = 1.0 / S[keep]
    tmp = U.T @ np.asarray(b, dtype=float)
    tmp = tmp * S_inv if tmp.ndim == 1 else S_inv[:, None] * tmp
    return Vt.T @ tmp
