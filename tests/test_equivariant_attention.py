"""SE(3)-equivariance of the equivariant transformer and end-to-end
training with it (pattern: reference tests/test_equivariant_*.py)."""

import numpy as np
import pytest
import torch

from hydragnn_amd.globalatt.equivariant import (
    EquivariantAllToAllAttention,
    EquivariantRMSNorm,
    EquivariantTransformerLayer,
    complete_graph_edge_index,
)
from hydragnn_amd.models.mace.o3 import dim
from _training_workflow import evaluate_error, run_training


def _rand_rot(seed=0):
    rng = np.random.default_rng(seed)
    Q, _ = np.linalg.qr(rng.normal(size=(3, 3)))
    if np.linalg.det(Q) < 0:
        Q[:, 0] *= -1
    return torch.from_numpy(Q)


def _wigner_block(R, lmax):
    """Block-diag Wigner D over the tower, from SH sampling."""
    from hydragnn_amd.ops import spherical_harmonics
    D = torch.zeros(dim(lmax), dim(lmax), dtype=torch.float64)
    rng = np.random.default_rng(3)
    v = torch.from_numpy(rng.normal(size=(200, 3)))
    for l in range(lmax + 1):
        sl = slice(l * l, (l + 1) ** 2)
        Y = spherical_harmonics(v, l)[:, sl]
        Yr = spherical_harmonics(v @ R.t(), l)[:, sl]
        sol = torch.linalg.lstsq(Y, Yr).solution
        D[sl, sl] = sol.t()
    return D


def test_complete_graph():
    batch = torch.tensor([0, 0, 0, 1, 1])
    ei = complete_graph_edge_index(batch)
    assert ei.shape[1] == 9 + 4
    assert (batch[ei[0]] == batch[ei[1]]).all()


@pytest.mark.parametrize("module", ["attention", "layer", "norm"])
def test_equivariance(module):
    torch.manual_seed(0)
    lmax, C, n = 2, 8, 12
    if module == "attention":
        mod = EquivariantAllToAllAttention(C, lmax, num_heads=2,
                                           chunk_size=None).double()
        fn = lambda f, p, b: mod(f, p, b)
    elif module == "layer":
        mod = EquivariantTransformerLayer(C, lmax, num_heads=2).double()
        fn = lambda f, p, b: mod(f, p, b)
    else:
        mod = EquivariantRMSNorm(C, lmax).double()
        fn = lambda f, p, b: mod(f)
    mod.eval()

    feats = torch.randn(n, C, dim(lmax), dtype=torch.float64)
    pos = torch.randn(n, 3, dtype=torch.float64)
    batch = torch.tensor([0] * 6 + [1] * 6)
    R = _rand_rot(1)
    Dw = _wigner_block(R, lmax)

    out1 = fn(torch.einsum("ij,ncj->nci", Dw, feats), pos @ R.t(), batch)
    out2 = torch.einsum("ij,ncj->nci", Dw, fn(feats, pos, batch))
    assert torch.allclose(out1, out2, atol=1e-8), (
        f"{module}: {(out1 - out2).abs().max():.2e}")


def test_translation_invariance():
    torch.manual_seed(0)
    lmax, C, n = 1, 8, 10
    mod = EquivariantAllToAllAttention(C, lmax, num_heads=2,
                                       chunk_size=None).double()
    mod.eval()
    feats = torch.randn(n, C, dim(lmax), dtype=torch.float64)
    pos = torch.randn(n, 3, dtype=torch.float64)
    batch = torch.zeros(n, dtype=torch.long)
    o1 = mod(feats, pos, batch)
    o2 = mod(feats, pos + 5.0, batch)
    assert torch.allclose(o1, o2, atol=1e-9)


def test_chunked_matches_full():
    torch.manual_seed(0)
    lmax, C, n = 1, 8, 30
    full = EquivariantAllToAllAttention(C, lmax, num_heads=2,
                                        chunk_size=None).double()
    chunked = EquivariantAllToAllAttention(C, lmax, num_heads=2,
                                           chunk_size=5).double()
    chunked.load_state_dict(full.state_dict())
    full.eval(); chunked.eval()
    feats = torch.randn(n, C, dim(lmax), dtype=torch.float64)
    pos = torch.randn(n, 3, dtype=torch.float64)
    batch = torch.zeros(n, dtype=torch.long)
    assert torch.allclose(full(feats, pos, batch),
                          chunked(feats, pos, batch), atol=1e-10)


def test_painn_with_equivariant_transformer_trains():
    overrides = {"NeuralNetwork": {"Architecture": {
        "global_attn_engine": "EquivariantTransformer",
        "global_attn_heads": 2,
        "equivariant_attn_lmax": 1,
        "equivariant_attn_num_radial": 8,
        "equivariant_attn_chunk_size": 512,
    }}}
    model, config, loaders = run_training(
        "PAINN", heads=("graph",), num_samples=32, num_epoch=10,
        overrides=overrides)
    err, rmses = evaluate_error(model, loaders[2], config)
    assert rmses[0] < 0.6, f"PAINN+ET RMSE {rmses[0]:.3f}"


@pytest.mark.parametrize("mpnn_type", ["PNAEq", "MACE", "EGNN"])
def test_equivariant_transformer_integrations(mpnn_type):
    """Reference test_equivariant_{pnaeq,mace,scalar_mpnn}_integration
    pattern: each family trains with the equivariant global attention
    engine and reaches its accuracy gate."""
    overrides = {"NeuralNetwork": {"Architecture": {
        "global_attn_engine": "EquivariantTransformer",
        "global_attn_heads": 2,
        "equivariant_attn_lmax": 1,
        "equivariant_attn_num_radial": 8,
        "equivariant_attn_chunk_size": 512,
    }}}
    if mpnn_type == "MACE":
        overrides["NeuralNetwork"]["Architecture"].update(
            {"max_ell": 2, "node_max_ell": 1, "correlation": 2,
             "num_radial": 8})
    if mpnn_type == "EGNN":
        # scalar-only coupling (reference scalar_mpnn integration)
        overrides["NeuralNetwork"]["Architecture"][
            "equivariant_attn_allow_scalar_only"] = True
    epochs = 25 if mpnn_type == "PNAEq" else 10
    model, config, loaders = run_training(
        mpnn_type, heads=("graph",), num_samples=48, num_epoch=epochs,
        overrides=overrides)
    err, rmses = evaluate_error(model, loaders[2], config)
    assert rmses[0] < 0.8, f"{mpnn_type}+ET RMSE {rmses[0]:.3f}"
