"""Correctness of the real-O(3) toolkit: SH basis consistency,
Wigner-3j equivariance, Gaunt product property."""

import math

import numpy as np
import pytest
import torch

from hydragnn_amd.models.mace import o3
from hydragnn_amd.ops import spherical_harmonics


def _complex_sh(l, v):
    """scipy complex SH evaluated at cartesian unit vectors, m=-l..l."""
    try:
        from scipy.special import sph_harm_y

        def _sh(m, l_, phi, theta):
            return sph_harm_y(l_, m, theta, phi)
    except ImportError:  # scipy < 1.15
        from scipy.special import sph_harm

        def _sh(m, l_, phi, theta):
            return sph_harm(m, l_, phi, theta)
    x, y, z = v[:, 0], v[:, 1], v[:, 2]
    theta = np.arccos(np.clip(z, -1, 1))       # polar
    phi = np.arctan2(y, x)                     # azimuth
    out = np.zeros((v.shape[0], 2 * l + 1), dtype=np.complex128)
    for m in range(-l, l + 1):
        out[:, m + l] = _sh(m, l, phi, theta)
    return out


@pytest.mark.parametrize("l", [0, 1, 2, 3])
def test_real_sh_matches_complex_basis(l):
    """ops.spherical_harmonics == sqrt(4pi) * U_l @ complex SH."""
    rng = np.random.default_rng(0)
    v = rng.normal(size=(50, 3))
    v /= np.linalg.norm(v, axis=1, keepdims=True)
    Yc = _complex_sh(l, v)
    U = o3._u_matrix(l)
    Yr = (Yc @ U.T) * math.sqrt(4 * math.pi)
    assert np.abs(Yr.imag).max() < 1e-10
    mine = spherical_harmonics(torch.from_numpy(v), 3).numpy()
    sl = slice(l * l, (l + 1) * (l + 1))
    assert np.abs(mine[:, sl] - Yr.real).max() < 1e-6


def _wigner_d(l, R):
    """D_l(R) from SH sampling: Y(Rv) = D Y(v), least squares."""
    rng = np.random.default_rng(3)
    v = rng.normal(size=(200, 3))
    v /= np.linalg.norm(v, axis=1, keepdims=True)
    Y = spherical_harmonics(torch.from_numpy(v), l).numpy()
    Yr = spherical_harmonics(torch.from_numpy(v @ R.T), l).numpy()
    sl = slice(l * l, (l + 1) * (l + 1))
    D, *_ = np.linalg.lstsq(Y[:, sl], Yr[:, sl], rcond=None)
    return D.T  # Y(Rv) = D @ Y(v)


def _rand_rot(seed=0):
    rng = np.random.default_rng(seed)
    A = rng.normal(size=(3, 3))
    Q, _ = np.linalg.qr(A)
    if np.linalg.det(Q) < 0:
        Q[:, 0] *= -1
    return Q


@pytest.mark.parametrize("lll", [(1, 1, 2), (1, 1, 0), (2, 1, 1),
                                 (2, 2, 2), (3, 2, 1), (2, 1, 3)])
def test_wigner3j_equivariance(lll):
    l1, l2, l3 = lll
    W = o3.wigner_3j(l1, l2, l3).numpy()
    assert np.abs(W).max() > 1e-6, "3j tensor should be nonzero"
    R = _rand_rot(5)
    D1, D2, D3 = _wigner_d(l1, R), _wigner_d(l2, R), _wigner_d(l3, R)
    W_rot = np.einsum("am,bn,co,mno->abc", D1, D2, D3, W)
    assert np.abs(W_rot - W).max() < 1e-5, (
        f"3j({lll}) not invariant: {np.abs(W_rot - W).max():.2e}")


def test_gaunt_product_property():
    """contract W3j with Y_l1(v) Y_l2(v) -> proportional to Y_l3(v)."""
    torch.manual_seed(0)
    v = torch.randn(40, 3)
    Y = spherical_harmonics(v, 3)
    for (l1, l2, l3) in [(1, 1, 2), (2, 1, 1), (1, 1, 0)]:
        W = o3.wigner_3j(l1, l2, l3).float()
        s1 = slice(l1 * l1, (l1 + 1) ** 2)
        s2 = slice(l2 * l2, (l2 + 1) ** 2)
        s3 = slice(l3 * l3, (l3 + 1) ** 2)
        prod = torch.einsum("abc,na,nb->nc", W, Y[:, s1], Y[:, s2])
        tgt = Y[:, s3]
        # proportionality: prod = alpha * tgt for a single alpha
        alpha = (prod * tgt).sum() / (tgt * tgt).sum()
        assert torch.allclose(prod, alpha * tgt, atol=1e-4), (l1, l2, l3)


def test_irreps_linear_equivariance():
    torch.manual_seed(0)
    lin = o3.IrrepsLinear(8, 4, lmax=2).double()
    x = torch.randn(10, 8, 9, dtype=torch.float64)
    R = _rand_rot(7)
    D = np.zeros((9, 9))
    for l in range(3):
        sl = slice(l * l, (l + 1) ** 2)
        D[sl, sl] = _wigner_d(l, R)
    Dt = torch.from_numpy(D)
    out1 = lin(torch.einsum("ij,ncj->nci", Dt, x))
    out2 = torch.einsum("ij,ncj->nci", Dt, lin(x))
    assert torch.allclose(out1, out2, atol=1e-8)
