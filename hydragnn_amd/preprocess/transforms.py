"""Graph feature transforms: Laplacian positional encodings, edge
lengths, rotation normalization.

Covers the reference's use of PyG transforms
(AddLaplacianEigenvectorPE in examples/md17/md17_mlip.py, Distance /
Spherical / PointPairFeatures and NormalizeRotation in
abstractrawdataset.py:346-402).
"""

from __future__ import annotations

import numpy as np
import torch

from ..data import Data
from ..ops import get_edge_vectors_and_lengths


def add_laplacian_pe(data: Data, k: int) -> Data:
    """k nontrivial eigenvectors of the symmetric-normalized Laplacian
    as data.pe [N, k]; sign-fixed deterministically."""
    n = data.num_nodes
    ei = data.edge_index
    A = np.zeros((n, n))
    if ei.numel() > 0:
        A[ei[0].numpy(), ei[1].numpy()] = 1.0
        A = np.maximum(A, A.T)
    d = A.sum(1)
    d_inv_sqrt = 1.0 / np.sqrt(np.maximum(d, 1e-12))
    L = np.eye(n) - (A * d_inv_sqrt[:, None]) * d_inv_sqrt[None, :]
    w, v = np.linalg.eigh(L)
    order = np.argsort(w)
    vecs = v[:, order[1:k + 1]]
    if vecs.shape[1] < k:
        vecs = np.pad(vecs, ((0, 0), (0, k - vecs.shape[1])))
    # deterministic sign: first nonzero entry positive
    for j in range(vecs.shape[1]):
        col = vecs[:, j]
        nz = np.flatnonzero(np.abs(col) > 1e-8)
        if nz.size and col[nz[0]] < 0:
            vecs[:, j] = -col
    data.pe = torch.from_numpy(vecs).float()
    src, dst = ei[0], ei[1]
    data.rel_pe = (data.pe[dst] - data.pe[src]).abs()
    return data


def add_edge_lengths(data: Data, max_length: float = 1.0) -> Data:
    """PyG Distance-transform equivalent: normalized edge length as
    edge_attr (reference abstractrawdataset.py:375-396)."""
    _, lengths = get_edge_vectors_and_lengths(
        data.pos, data.edge_index, data.get("edge_shifts"))
    attr = (lengths / max_length).to(torch.float32)
    existing = data.get("edge_attr")
    if existing is not None:
        data.edge_attr = torch.cat([existing, attr], dim=-1)
    else:
        data.edge_attr = attr
    return data


def normalize_rotation(data: Data) -> Data:
    """Rotate positions into the PCA eigenbasis (PyG NormalizeRotation
    equivalent; reference abstractrawdataset.py:346)."""
    pos = data.pos - data.pos.mean(dim=0, keepdim=True)
    cov = pos.t() @ pos
    _, v = torch.linalg.eigh(cov.double())
    v = v.flip(-1)  # largest first
    # fix signs deterministically
    for j in range(3):
        col = v[:, j]
        nz = (col.abs() > 1e-8).nonzero()
        if nz.numel() and col[nz[0, 0]] < 0:
            v[:, j] = -col
    if torch.det(v) < 0:
        v[:, -1] = -v[:, -1]
    data.pos = (pos.double() @ v).to(pos.dtype)
    if data.get("forces") is not None:
        data.forces = (data.forces.double() @ v).to(data.forces.dtype)
    return data


def pbc_as_tensor(pbc) -> "torch.Tensor":
    """Normalize a pbc spec (bool | sequence of 3) to a bool [3] tensor
    (reference graph_samples_checks_and_updates.py:506)."""
    if isinstance(pbc, bool):
        return torch.tensor([pbc] * 3)
    t = torch.as_tensor(pbc).flatten().bool()
    assert t.numel() == 3, "pbc must have 3 components"
    return t


def pbc_distance(data, norm: bool = False, max_length: float = 1.0):
    """PBC-aware Distance transform: edge lengths using edge_shifts
    (reference PBCDistance, graph_samples:439)."""
    return add_edge_lengths(data, max_length=max_length if norm else 1.0)


def pbc_local_cartesian(data):
    """PBC-aware LocalCartesian: shift-corrected edge vectors as edge
    attrs (reference PBCLocalCartesian, graph_samples:470)."""
    vec, _ = get_edge_vectors_and_lengths(
        data.pos, data.edge_index, data.get("edge_shifts"))
    existing = data.get("edge_attr")
    vec = vec.to(torch.float32)
    data.edge_attr = torch.cat([existing, vec], dim=-1) \
        if existing is not None else vec
    return data
