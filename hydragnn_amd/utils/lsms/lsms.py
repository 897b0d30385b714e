"""LSMS post-processing utilities (reference: hydragnn/utils/lsms/*,
~268 LoC): formation-enthalpy / Gibbs free-energy conversion of raw
total energies for binary alloys, compositional histogram cutoff."""

from __future__ import annotations

from typing import Dict, List, Sequence

import numpy as np
import torch


def get_formation_enthalpy(total_energy: float, composition: Dict[int, int],
                           pure_energies: Dict[int, float]) -> float:
    """E_form = E_total - sum_z n_z * E_pure(z) / N_pure."""
    e = float(total_energy)
    for z, n in composition.items():
        e -= n * pure_energies[z]
    return e


def convert_raw_data_energy_to_gibbs(dataset: Sequence,
                                     pure_energies: Dict[int, float],
                                     temperature: float = 0.0) -> None:
    """Replace each sample's total energy by the formation enthalpy
    (plus an ideal-mixing entropy term at finite temperature)."""
    kB = 8.617333262e-5  # eV/K
    for d in dataset:
        z = d.get("z")
        if z is None:
            z = d.x[:, 0].long()
        z = z.flatten()
        comp = {int(v): int(c) for v, c in
                zip(*torch.unique(z, return_counts=True))}
        n = int(z.numel())
        e_form = get_formation_enthalpy(float(d.y.flatten()[0]), comp,
                                        pure_energies)
        if temperature > 0 and len(comp) > 1:
            xs = np.array([c / n for c in comp.values()])
            entropy = -kB * n * float((xs * np.log(xs)).sum())
            e_form = e_form - temperature * entropy
        d.y = torch.tensor([[e_form]], dtype=d.y.dtype)


def compositional_histogram_cutoff(dataset: Sequence, element: int,
                                   num_bins: int = 100,
                                   max_per_bin: int = 1000) -> List:
    """Cap the number of samples per composition bin of `element`
    (balances strongly peaked composition histograms)."""
    bins: Dict[int, int] = {}
    kept = []
    for d in dataset:
        z = d.get("z")
        if z is None:
            z = d.x[:, 0].long()
        z = z.flatten()
        frac = float((z == element).sum()) / max(z.numel(), 1)
        b = min(int(frac * num_bins), num_bins - 1)
        if bins.get(b, 0) < max_per_bin:
            bins[b] = bins.get(b, 0) + 1
            kept.append(d)
    return kept


# reference-named aliases (convert_total_energy_to_formation_gibbs.py)
compute_formation_enthalpy = get_formation_enthalpy


def read_file(path: str):
    """Read one raw LSMS text sample -> (free_energy, atom_rows)
    (reference convert_total_energy_to_formation_gibbs.read_file)."""
    with open(path) as f:
        lines = [ln.split() for ln in f.read().splitlines() if ln.strip()]
    free_energy = float(lines[0][0])
    atoms = [[float(v) for v in row] for row in lines[1:]]
    return free_energy, atoms


def find_bin(value: float, edges) -> int:
    """Histogram bin index for the compositional cutoff (reference
    compositional_histogram_cutoff.find_bin)."""
    for i in range(len(edges) - 1):
        if edges[i] <= value < edges[i + 1]:
            return i
    return len(edges) - 2


# reference-named alias
compute_formation_enthalpy = get_formation_enthalpy
