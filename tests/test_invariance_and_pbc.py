"""Rotational invariance of graph construction + PBC correctness
(patterns: reference tests/test_rotational_invariance.py:52-110,
test_periodic_boundary_conditions.py:82-101)."""

import numpy as np
import pytest
import torch

from hydragnn_amd.ops import (
    get_edge_vectors_and_lengths,
    radius_graph,
    radius_graph_pbc,
)
from hydragnn_amd.preprocess import normalize_rotation
from hydragnn_amd.data import Data


def _rand_rot(seed):
    rng = np.random.default_rng(seed)
    Q, _ = np.linalg.qr(rng.normal(size=(3, 3)))
    if np.linalg.det(Q) < 0:
        Q[:, 0] *= -1
    return torch.from_numpy(Q)


def test_radius_graph_rotation_invariant():
    """Edge set and edge lengths are invariant under rotation."""
    torch.manual_seed(0)
    pos = torch.rand(40, 3, dtype=torch.float64) * 3
    R = _rand_rot(1)
    ei1 = radius_graph(pos, 1.2, max_num_neighbors=1000)
    ei2 = radius_graph(pos @ R.t(), 1.2, max_num_neighbors=1000)
    s1 = {(int(a), int(b)) for a, b in ei1.t().tolist()}
    s2 = {(int(a), int(b)) for a, b in ei2.t().tolist()}
    assert s1 == s2
    _, l1 = get_edge_vectors_and_lengths(pos, ei1)
    _, l2 = get_edge_vectors_and_lengths(pos @ R.t(), ei1)
    assert torch.allclose(l1, l2, atol=1e-12)


def test_normalize_rotation_deterministic():
    torch.manual_seed(0)
    pos = torch.rand(20, 3, dtype=torch.float64)
    d1 = Data(pos=pos.clone(), x=torch.ones(20, 1))
    d2 = Data(pos=pos.clone(), x=torch.ones(20, 1))
    normalize_rotation(d1)
    normalize_rotation(d2)
    assert torch.allclose(d1.pos, d2.pos, atol=1e-14)
    # pairwise distances preserved
    assert torch.allclose(torch.cdist(d1.pos, d1.pos),
                          torch.cdist(pos - pos.mean(0), pos - pos.mean(0)),
                          atol=1e-10)


def test_pbc_h2_molecule():
    """Two atoms straddling the cell boundary: the PBC graph must find
    the short periodic bond (reference test pattern: H2 across the
    boundary)."""
    cell = torch.eye(3, dtype=torch.float64) * 10.0
    pos = torch.tensor([[0.3, 5.0, 5.0], [9.7, 5.0, 5.0]],
                       dtype=torch.float64)
    ei, shifts = radius_graph_pbc(pos, 1.0, cell)
    assert ei.shape[1] == 2  # one bond in each direction
    vec, lengths = get_edge_vectors_and_lengths(pos, ei, shifts)
    assert torch.allclose(lengths, torch.full_like(lengths, 0.6),
                          atol=1e-10)


def test_pbc_large_cutoff_multiple_images():
    """Cutoff larger than the box: neighbors include multiple periodic
    images of the same atom."""
    cell = torch.eye(3, dtype=torch.float64) * 2.0
    pos = torch.tensor([[0.0, 0.0, 0.0], [1.0, 1.0, 1.0]],
                       dtype=torch.float64)
    ei, shifts = radius_graph_pbc(pos, 2.5, cell)
    # atom 0 sees many images of atom 1 and of itself
    assert ei.shape[1] > 10
    vec, lengths = get_edge_vectors_and_lengths(pos, ei, shifts)
    assert (lengths <= 2.5 + 1e-9).all()
    assert (lengths > 1e-9).all()


def test_pbc_mixed_boundaries():
    cell = torch.eye(3, dtype=torch.float64) * 4.0
    pos = torch.tensor([[0.2, 2.0, 2.0], [3.8, 2.0, 2.0]],
                       dtype=torch.float64)
    # periodic only in y/z: the short x-wrap bond must NOT appear
    ei, shifts = radius_graph_pbc(pos, 1.0, cell,
                                  pbc=(False, True, True))
    assert ei.shape[1] == 0
    ei2, _ = radius_graph_pbc(pos, 1.0, cell, pbc=(True, True, True))
    assert ei2.shape[1] == 2


def test_pbc_lj_forces_match_open_in_big_box():
    """With a box much larger than the cutoff, PBC and open-boundary
    energies/forces agree."""
    from hydragnn_amd.utils.datasets.synthetic import (
        _lj_energy_forces)
    torch.manual_seed(2)
    pos = (torch.rand(10, 3, dtype=torch.float64) * 3) + 3.0
    cell = torch.eye(3, dtype=torch.float64) * 50.0
    ei_o = radius_graph(pos, 2.5, max_num_neighbors=100)
    ei_p, sh = radius_graph_pbc(pos, 2.5, cell)
    e1, f1 = _lj_energy_forces(pos, ei_o, None, 0.01, 1.0)
    e2, f2 = _lj_energy_forces(pos, ei_p, sh, 0.01, 1.0)
    assert torch.allclose(e1, e2, atol=1e-10)
    assert torch.allclose(f1, f2, atol=1e-10)


def test_rotated_sample_equivalence_after_normalization():
    """Reference test_rotational_invariance pattern: rotate a
    structure, rebuild edges, NormalizeRotation both — the samples are
    equivalent (edge-order-insensitive) to 1e-14-grade tolerance."""
    from hydragnn_amd.preprocess import (check_data_samples_equivalence,
                                         get_radius_graph)
    torch.manual_seed(0)
    pos = torch.rand(24, 3, dtype=torch.float64)
    q, _ = torch.linalg.qr(torch.randn(3, 3, dtype=torch.float64))
    if torch.det(q) < 0:
        q[:, 0] = -q[:, 0]

    def build(p):
        d = Data(pos=p.clone(), x=torch.ones(24, 1, dtype=torch.float64),
                 y=torch.zeros(1, dtype=torch.float64))
        normalize_rotation(d)
        get_radius_graph(0.6, 32)(d)
        src, dst = d.edge_index[0], d.edge_index[1]
        d.edge_attr = (d.pos[src] - d.pos[dst]).norm(dim=-1,
                                                     keepdim=True)
        return d

    d1 = build(pos)
    d2 = build(pos @ q.t())
    assert check_data_samples_equivalence(d1, d2, 1e-12)
