"""MACE symmetric contraction (n-body product basis, Eqs. 10-11).

Replaces e3nn/opt_einsum_fx U-matrix machinery
(/root/reference/hydragnn/utils/model/mace_utils/modules/
symmetric_contraction.py:39-249 + tools/cg.py): U^(nu) basis tensors of
symmetric equivariant couplings of nu copies of the SH-like feature
tower to each output l, built at init from our Wigner-3j cache
(Gram-Schmidt over symmetrized coupling paths), contracted with
per-element weights by the progressive einsum scheme.
"""

from __future__ import annotations

import itertools
import math
from functools import lru_cache
from typing import List

import numpy as np
import torch
from torch import nn

from .o3 import allowed_l3, dim, wigner_3j


def _embed2(l1: int, l2: int, lout: int, lmax: int) -> np.ndarray:
    """W3j(l1,l2,lout) embedded into full tower slots:
    [2lout+1, D, D]."""
    D = dim(lmax)
    W = wigner_3j(l1, l2, lout).numpy()  # [2l1+1, 2l2+1, 2lout+1]
    out = np.zeros((2 * lout + 1, D, D))
    s1, s2 = l1 * l1, l2 * l2
    out[:, s1:s1 + 2 * l1 + 1, s2:s2 + 2 * l2 + 1] = \
        np.transpose(W, (2, 0, 1))
    return out


def _symmetrize(T: np.ndarray) -> np.ndarray:
    """Symmetrize over the nu slot axes (axes 1..nu)."""
    nu = T.ndim - 1
    acc = np.zeros_like(T)
    for perm in itertools.permutations(range(nu)):
        axes = (0,) + tuple(1 + p for p in perm)
        acc += np.transpose(T, axes)
    return acc / math.factorial(nu)


def _gram_schmidt(tensors: List[np.ndarray], tol: float = 1e-9
                  ) -> List[np.ndarray]:
    basis = []
    for T in tensors:
        v = T.reshape(-1).copy()
        for b in basis:
            v -= (v @ b.reshape(-1)) * b.reshape(-1).reshape(v.shape)
        n = np.linalg.norm(v)
        if n > tol:
            basis.append((v / n).reshape(T.shape))
    return basis


@lru_cache(maxsize=None)
def u_matrix(lmax: int, lout: int, nu: int):
    """U^(nu)_{lout}: [2lout+1, D^nu..., K] orthonormal symmetric
    coupling basis (numpy, cached)."""
    D = dim(lmax)
    cands: List[np.ndarray] = []
    if nu == 1:
        if lout <= lmax:
            T = np.zeros((2 * lout + 1, D))
            s = lout * lout
            T[:, s:s + 2 * lout + 1] = np.eye(2 * lout + 1)
            cands.append(T)
    elif nu == 2:
        for l1 in range(lmax + 1):
            for l2 in range(lmax + 1):
                if lout in allowed_l3(l1, l2, lout):
                    cands.append(_symmetrize(_embed2(l1, l2, lout, lmax)))
    elif nu == 3:
        for l1 in range(lmax + 1):
            for l2 in range(lmax + 1):
                for l12 in allowed_l3(l1, l2, 2 * lmax):
                    W12 = wigner_3j(l1, l2, l12).numpy()
                    for l3 in range(lmax + 1):
                        if lout not in allowed_l3(l12, l3, lout):
                            continue
                        W3 = wigner_3j(l12, l3, lout).numpy()
                        T = np.einsum("abm,mco->oabc", W12, W3)
                        full = np.zeros((2 * lout + 1, D, D, D))
                        s1, s2, s3 = l1 * l1, l2 * l2, l3 * l3
                        full[:, s1:s1 + 2 * l1 + 1, s2:s2 + 2 * l2 + 1,
                             s3:s3 + 2 * l3 + 1] = T
                        cands.append(_symmetrize(full))
    else:
        raise NotImplementedError(f"correlation {nu} > 3 not supported")
    basis = _gram_schmidt(cands)
    if not basis:
        shape = (2 * lout + 1,) + (D,) * nu + (0,)
        return torch.zeros(shape)
    U = np.stack(basis, axis=-1)
    return torch.from_numpy(np.ascontiguousarray(U)).float()


class Contraction(nn.Module):
    """Symmetric contraction to one output l: the progressive
    (highest-nu first) contraction scheme of MACE."""

    def __init__(self, lmax: int, lout: int, correlation: int,
                 num_channels: int, num_elements: int):
        super().__init__()
        self.lmax = lmax
        self.lout = lout
        self.correlation = correlation
        self.weights = nn.ParameterDict()
        self.us = {}
        for nu in range(correlation, 0, -1):
            U = u_matrix(lmax, lout, nu)
            self.register_buffer(f"U{nu}", U)
            k = U.shape[-1]
            w = nn.Parameter(
                torch.randn(num_elements, k, num_channels) / max(k, 1))
            self.weights[str(nu)] = w

    def _cu(self, nu_key: int, node_elem, x_dtype):
        """U[m] . w  ->  [n, c, o, i1..i_m] via ONE dense GEMM
        (M = n*c, K = num_paths, N = flat) instead of tiny-batched
        einsums."""
        from ...ops import gather
        U = getattr(self, f"U{nu_key}").to(x_dtype)
        wfull = self.weights[str(nu_key)].to(x_dtype)
        nel, k, c = wfull.shape
        # per-node element gather via our op (backward = fused
        # scatter-add instead of torch's indexing_backward kernel)
        w = gather(wfull.reshape(nel, k * c), node_elem).view(-1, k, c)
        n = w.shape[0]
        flat = U.reshape(-1, max(k, 1))  # [o*D^m, k]
        wp = w.permute(0, 2, 1).reshape(n * c, k)
        cu = wp @ flat.t()  # [n*c, o*D^m]
        return cu.reshape((n, c) + U.shape[:-1])

    def forward(self, x: torch.Tensor, node_elem: torch.Tensor
                ) -> torch.Tensor:
        """x [N, C, D]; node_elem [N] element ids -> [N, C, 2lout+1]."""
        from ...ops.etp import fold_last
        nu = self.correlation
        U = getattr(self, f"U{nu}")
        if U.shape[-1] == 0:
            D = x.shape[-1]
            out = x.new_zeros((x.shape[0], x.shape[1],
                               2 * self.lout + 1) + (D,) * (nu - 1))
        else:
            cu = self._cu(nu, node_elem, x.dtype)
            out = fold_last(cu, x)  # [n, c, o, i1..i_{nu-1}]
        for m in range(nu - 1, 0, -1):
            Um = getattr(self, f"U{m}")
            if Um.shape[-1] > 0:
                out = self._cu(m, node_elem, x.dtype) + out
            out = fold_last(out, x)
        return out  # [N, C, 2lout+1]


class SymmetricContraction(nn.Module):
    def __init__(self, lmax_in: int, lmax_out: int, correlation: int,
                 num_channels: int, num_elements: int):
        super().__init__()
        self.contractions = nn.ModuleList([
            Contraction(lmax_in, lout, correlation, num_channels,
                        num_elements)
            for lout in range(lmax_out + 1)])

    def forward(self, x: torch.Tensor, node_elem: torch.Tensor
                ) -> torch.Tensor:
        outs = [c(x, node_elem) for c in self.contractions]
        return torch.cat(outs, dim=-1)  # [N, C, dim(lmax_out)]
