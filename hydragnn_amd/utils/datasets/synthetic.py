"""Synthetic datasets with closed-form targets.

The MLIP analogue of the reference's Lennard-Jones generator
(/root/reference/examples/LennardJones/LJ_data.py:53-450): random
configurations with analytic LJ energies and forces (open or periodic
boundary), used for force-training tests and the bench's MD17-shape
synthetic molecules (no network access — datasets are generated).
"""

from __future__ import annotations

import math
from typing import Optional, Sequence

import torch

from ...data import Data
from ...ops import radius_graph, radius_graph_pbc, scatter


def _lj_energy_forces(pos: torch.Tensor, edge_index: torch.Tensor,
                      shifts: Optional[torch.Tensor], epsilon: float,
                      sigma: float, r_min: float = 0.0):
    """Pairwise LJ over the given edge list (each pair appears twice —
    once per direction — so use 0.5x for energy).  ``r_min`` soft-cores
    the potential: distances below it are clamped in the energy/force
    evaluation so near-coincident atoms (random geometries) cannot
    produce astronomically large targets."""
    src, dst = edge_index[0], edge_index[1]
    vec = pos[dst] - pos[src]
    if shifts is not None:
        vec = vec + shifts
    r2 = (vec * vec).sum(-1).clamp(min=max(1e-12, r_min * r_min))
    inv_r2 = (sigma * sigma) / r2
    inv_r6 = inv_r2 ** 3
    inv_r12 = inv_r6 ** 2
    e_pair = 4.0 * epsilon * (inv_r12 - inv_r6)
    energy = 0.5 * e_pair.sum()
    # dE/dr_ij along vec: f = 24 eps (2 r^-12 - r^-6) / r^2 * vec
    coef = 24.0 * epsilon * (2.0 * inv_r12 - inv_r6) / r2
    f_edge = coef.unsqueeze(-1) * vec
    # force on dst from src along +vec, on src along -vec; sum halves
    n = pos.shape[0]
    forces = scatter(f_edge, dst, n, "sum")
    return energy, forces


def lj_dataset(
    num_samples: int = 32,
    num_atoms: int = 32,
    cell_size: float = 6.0,
    radius: float = 2.5,
    pbc: bool = True,
    epsilon: float = 0.01,
    sigma: float = 1.0,
    seed: int = 11,
    dtype: torch.dtype = torch.float32,
):
    """Random near-lattice LJ configurations. x = [Z]; energy/forces
    analytic; edges from the PBC-aware radius graph with shift vectors."""
    g = torch.Generator().manual_seed(seed)
    n_side = max(1, math.ceil(num_atoms ** (1 / 3) - 1e-9))
    spacing = cell_size / n_side
    base = torch.stack(torch.meshgrid(
        torch.arange(n_side), torch.arange(n_side), torch.arange(n_side),
        indexing="ij"), dim=-1).reshape(-1, 3).to(dtype) * spacing
    base = base[:num_atoms]
    n = base.shape[0]
    cell = torch.eye(3, dtype=dtype) * cell_size

    dataset = []
    for _ in range(num_samples):
        pos = base + (torch.rand(n, 3, generator=g) - 0.5) * 0.2 * spacing
        pos = pos.to(dtype)
        if pbc:
            edge_index, shifts = radius_graph_pbc(pos, radius, cell)
            shifts = shifts.to(dtype)
        else:
            edge_index = radius_graph(pos, radius, max_num_neighbors=1000)
            shifts = None
        energy, forces = _lj_energy_forces(pos.double(), edge_index,
                                           None if shifts is None
                                           else shifts.double(),
                                           epsilon, sigma)
        d = Data(
            x=torch.ones(n, 1, dtype=dtype),
            z=torch.full((n,), 13, dtype=torch.long),
            pos=pos,
            edge_index=edge_index,
            energy=energy.to(dtype).view(1, 1),
            forces=forces.to(dtype),
            y=energy.to(dtype).view(1, 1),
        )
        if shifts is not None:
            d.edge_shifts = shifts
            d.cell = cell.view(1, 3, 3)
        d.num_nodes = n
        dataset.append(d)
    return dataset


# MD17-like molecules: aspirin has 21 atoms (C9H8O4)
_MD17_SPECIES = [6] * 9 + [1] * 8 + [8] * 4


def md17_shape_dataset(
    num_samples: int = 64,
    radius: float = 7.0,
    max_neighbours: int = 30,
    seed: int = 13,
    dtype: torch.dtype = torch.float32,
    species: Sequence[int] = _MD17_SPECIES,
    spread: float = 2.5,
):
    """MD17-shaped molecules (aspirin atom count/species) with
    synthetic LJ-form energies/forces — the headline-bench data shape
    (BASELINE.json configs[1]); pre-transform mirrors
    examples/md17/md17_mlip.py:31-120: x = Z, y = energy/len(x),
    forces, radius graph r=7."""
    g = torch.Generator().manual_seed(seed)
    z = torch.tensor(list(species), dtype=torch.long)
    n = z.numel()
    dataset = []
    for _ in range(num_samples):
        pos = (torch.rand(n, 3, generator=g) - 0.5) * 2 * spread
        # push apart overlapping atoms to keep LJ finite
        for _ in range(3):
            d = torch.cdist(pos, pos) + torch.eye(n) * 10
            mind = d.min()
            if mind > 0.7:
                break
            pos = pos * 1.25
        pos = pos.to(dtype)
        edge_index = radius_graph(pos, radius,
                                  max_num_neighbors=max_neighbours)
        energy, forces = _lj_energy_forces(
            pos.double(), edge_index, None, 0.05, 1.0, r_min=0.7)
        d = Data(
            x=z.to(dtype).view(-1, 1),
            z=z.clone(),
            pos=pos,
            edge_index=edge_index,
            energy=energy.to(dtype).view(1, 1),
            forces=forces.to(dtype),
            y=(energy.to(dtype) / n).view(1, 1),
        )
        d.num_nodes = n
        dataset.append(d)
    return dataset


def md17_shape_dataset_fast(
    num_samples: int,
    radius: float = 7.0,
    seed: int = 13,
    dtype: torch.dtype = torch.float32,
    species: Sequence[int] = _MD17_SPECIES,
    spread: float = 2.5,
    chunk: int = 2048,
    min_dist: float = 0.7,
    max_push: int = 3,
):
    """Vectorized md17_shape_dataset: generates molecules in batched
    chunks (one cdist/LJ evaluation per chunk instead of per molecule)
    so bench-scale datasets (tens of thousands of samples) build in
    seconds.  Same distribution as md17_shape_dataset; edge lists are
    dst-major sorted (CSR-friendly).  No neighbor cap: aspirin-shape
    molecules have at most n-1=20 neighbours < the bench cap of 30."""
    g = torch.Generator().manual_seed(seed)
    z = torch.tensor(list(species), dtype=torch.long)
    n = z.numel()
    x_feat = z.to(dtype).view(-1, 1)
    eye = torch.eye(n, dtype=torch.bool)
    out = []
    for start in range(0, num_samples, chunk):
        m = min(chunk, num_samples - start)
        pos = (torch.rand(m, n, 3, generator=g) - 0.5) * 2 * spread
        for _ in range(max_push):
            d = torch.cdist(pos, pos) + eye * 10
            mind = d.flatten(1).min(dim=1).values
            scale = torch.where(mind > min_dist,
                                torch.ones(m), torch.full((m,), 1.25))
            if (scale == 1.0).all():
                break
            pos = pos * scale.view(-1, 1, 1)
        posd = pos.double()
        # _lj_energy_forces convention: vec = pos[dst] - pos[src];
        # vec[b, src, dst, :] = pos[b, dst] - pos[b, src]
        vec = posd.unsqueeze(1) - posd.unsqueeze(1).transpose(1, 2)
        r2 = (vec * vec).sum(-1).clamp(min=1e-12)
        within = (r2 < radius * radius) & ~eye
        # soft-core at min_dist: the 1.25x push loop cannot separate
        # near-coincident pairs (a multiplicative rescale leaves tiny
        # distances tiny), and an unclamped LJ at r~0.05 produces
        # ~1e15 force targets that blow the initial loss up to ~1e30
        # (observed tripping the captured-step sanity check).
        r2 = r2.clamp(min=min_dist * min_dist)
        sigma2 = 1.0
        inv_r2 = sigma2 / r2
        inv_r6 = inv_r2 ** 3
        inv_r12 = inv_r6 ** 2
        epsilon = 0.05
        e_pair = 4.0 * epsilon * (inv_r12 - inv_r6) * within
        energy = 0.5 * e_pair.sum(dim=(1, 2))
        coef = (24.0 * epsilon * (2.0 * inv_r12 - inv_r6) / r2) * within
        # force on dst: sum over src of coef * vec  -> reduce dim 1
        forces = (coef.unsqueeze(-1) * vec).sum(dim=1)
        # dst-major edge extraction: index mask as [dst, src]
        mask_ds = within.transpose(1, 2)  # [b, dst, src]
        nz = mask_ds.reshape(m, -1).nonzero(as_tuple=False)
        b_idx, flat = nz[:, 0], nz[:, 1]
        dsts, srcs = flat // n, flat % n
        counts = torch.bincount(b_idx, minlength=m)
        offs = torch.zeros(m + 1, dtype=torch.long)
        offs[1:] = counts.cumsum(0)
        pos_f = pos.to(dtype)
        energy_f = energy.to(dtype)
        forces_f = forces.to(dtype)
        for b in range(m):
            lo, hi = int(offs[b]), int(offs[b + 1])
            ei = torch.stack([srcs[lo:hi], dsts[lo:hi]], dim=0)
            d = Data(
                x=x_feat.clone(),
                z=z.clone(),
                pos=pos_f[b].clone(),
                edge_index=ei,
                energy=energy_f[b].view(1, 1).clone(),
                forces=forces_f[b].clone(),
                y=(energy_f[b] / n).view(1, 1).clone(),
            )
            d.num_nodes = n
            out.append(d)
    return out
