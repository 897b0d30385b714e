"""GPS global-attention layer tests (pattern: reference
tests/test_graphgps_pyg_parity.py — here vs our own eager reference and
end-to-end training)."""

import pytest
import torch

from _training_workflow import evaluate_error, run_training
from hydragnn_amd.data import Data
from hydragnn_amd.preprocess import add_laplacian_pe


def _with_pe(ds_kwargs=None):
    return dict(ds_kwargs or {})


@pytest.mark.parametrize("attn_type", ["multihead", "performer"])
def test_gps_training(attn_type):
    overrides = {"NeuralNetwork": {"Architecture": {
        "global_attn_engine": "gps",
        "global_attn_type": attn_type,
        "global_attn_heads": 4,
        "pe_dim": 3,
    }}}
    # patch the dataset: add laplacian PE
    import deterministic_graph_data as dgd
    orig = dgd.make_deterministic_dataset

    def with_pe(*a, **k):
        ds = orig(*a, **k)
        for d in ds:
            add_laplacian_pe(d, 3)
        return ds

    dgd.make_deterministic_dataset = with_pe
    import _training_workflow as tw
    tw.make_deterministic_dataset = with_pe
    try:
        model, config, loaders = run_training(
            "GIN", heads=("graph",), num_samples=48, num_epoch=20,
            overrides=overrides)
        err, rmses = evaluate_error(model, loaders[2], config)
        assert rmses[0] < 0.4, f"GPS({attn_type}) RMSE {rmses[0]:.3f}"
    finally:
        dgd.make_deterministic_dataset = orig
        tw.make_deterministic_dataset = orig


def test_performer_redraw():
    from hydragnn_amd.globalatt import PerformerAttention
    torch.manual_seed(0)
    attn = PerformerAttention(16, 4)
    p0 = attn.projection.clone()
    attn.redraw_projection_matrix()
    assert not torch.allclose(p0, attn.projection)


def test_laplacian_pe_deterministic():
    torch.manual_seed(0)
    from hydragnn_amd.ops import radius_graph
    pos = torch.rand(10, 3)
    d1 = Data(x=torch.rand(10, 1), pos=pos,
              edge_index=radius_graph(pos, 0.8))
    d2 = d1.clone()
    add_laplacian_pe(d1, 3)
    add_laplacian_pe(d2, 3)
    assert torch.allclose(d1.pe, d2.pe)
    assert d1.pe.shape == (10, 3)


def test_torch_varlen_attention_reference():
    """The varlen reference (used as the HIP kernel's numerics baseline
    and recompute backward) equals naive per-graph attention."""
    from hydragnn_amd.ops.varlen_attn import torch_varlen_attention
    torch.manual_seed(0)
    sizes = [3, 1, 9, 6]
    batch = torch.repeat_interleave(torch.arange(4), torch.tensor(sizes))
    N, H, dh = sum(sizes), 2, 8
    q, k, v = (torch.randn(N, H, dh) for _ in range(3))
    out = torch_varlen_attention(q, k, v, batch)
    lo = 0
    for n in sizes:
        qs = q[lo:lo + n].transpose(0, 1)  # [H, n, dh]
        ks = k[lo:lo + n].transpose(0, 1)
        vs = v[lo:lo + n].transpose(0, 1)
        a = torch.softmax(qs @ ks.transpose(-1, -2) / dh ** 0.5, dim=-1)
        ref = (a @ vs).transpose(0, 1)
        assert torch.allclose(out[lo:lo + n], ref, atol=1e-5)
        lo += n


def test_varlen_chunked_reference_matches_dense():
    """Chunked exact attention (large-segment recompute backward)
    equals the dense reference, values and gradients."""
    import math
    from hydragnn_amd.ops.varlen_attn import (
        torch_varlen_attention, torch_varlen_attention_chunked)
    torch.manual_seed(2)
    sizes = [700, 5, 130]
    batch = torch.repeat_interleave(torch.arange(3),
                                    torch.tensor(sizes))
    N, H, dh = sum(sizes), 2, 16
    q = torch.randn(N, H, dh, requires_grad=True)
    k = torch.randn(N, H, dh, requires_grad=True)
    v = torch.randn(N, H, dh)
    o1 = torch_varlen_attention(q, k, v, batch)
    o2 = torch_varlen_attention_chunked(q, k, v, batch, chunk=128)
    assert torch.allclose(o1, o2, atol=1e-5)
    g1 = torch.autograd.grad(o1.square().sum(), (q, k),
                             retain_graph=True)
    g2 = torch.autograd.grad(o2.square().sum(), (q, k))
    for a, b in zip(g1, g2):
        assert torch.allclose(a, b, atol=1e-4)
