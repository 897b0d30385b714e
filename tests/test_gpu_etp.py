"""Fused ETP kernel numerics vs the dense-einsum reference, including
first- and second-order gradients (the force-training path)."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("requires MI355X GPU", allow_module_level=True)

from hydragnn_amd.ops.etp import (  # noqa: E402
    ETPTable, _dense_general, etp_general, etp_reduce, fold_last,
)


def _table(seed=0, da=4, db=9, dg=6, do=16, n_ent=40):
    g = torch.Generator().manual_seed(seed)
    ents = torch.stack([
        torch.randint(0, da, (n_ent,), generator=g),
        torch.randint(0, db, (n_ent,), generator=g),
        torch.randint(0, dg, (n_ent,), generator=g),
        torch.randint(0, do, (n_ent,), generator=g),
    ], dim=1)
    coefs = torch.randn(n_ent, generator=g)
    return ETPTable(ents, coefs, (da, db, dg, do))


def test_etp_forward_matches_dense():
    torch.manual_seed(0)
    tab = _table()
    E, C = 500, 32
    A = torch.randn(E, C, 4, device="cuda")
    B = torch.randn(E, 9, device="cuda")
    Cw = torch.randn(E, C, 6, device="cuda")
    out = etp_general(A, B, Cw, tab)
    ref = _dense_general(A, B, Cw, tab)
    assert torch.allclose(out, ref, atol=1e-4), (
        (out - ref).abs().max().item())


def test_etp_first_and_second_grads():
    torch.manual_seed(1)
    tab = _table()
    E, C = 64, 8

    def run(dense):
        torch.manual_seed(2)
        A = torch.randn(E, C, 4, device="cuda", requires_grad=True)
        B = torch.randn(E, 9, device="cuda", requires_grad=True)
        Cw = torch.randn(E, C, 6, device="cuda", requires_grad=True)
        if dense:
            out = _dense_general(A, B, Cw, tab)
        else:
            out = etp_general(A, B, Cw, tab)
        loss = (out.float() ** 2).sum()
        gA, gB, gC = torch.autograd.grad(loss, (A, B, Cw),
                                         create_graph=True)
        # second order: grad of |gA|^2 wrt all inputs
        loss2 = (gA ** 2).sum() + (gB ** 2).sum() + (gC ** 2).sum()
        g2 = torch.autograd.grad(loss2, (A, B, Cw))
        return (gA, gB, gC) + tuple(g2)

    fused = run(False)
    dense = run(True)
    for i, (f, d) in enumerate(zip(fused, dense)):
        assert torch.allclose(f, d, atol=2e-3, rtol=1e-3), (
            f"grad {i} mismatch {(f - d).abs().max().item():.2e}")


def test_etp_reduce_matches_dense():
    torch.manual_seed(3)
    tab = _table()
    E, C = 300, 16
    A = torch.randn(E, C, 4, device="cuda")
    Cw = torch.randn(E, C, 6, device="cuda")
    D = torch.randn(E, C, 16, device="cuda")
    out = etp_reduce(A, Cw, D, tab)
    W = tab.dense(A.device, torch.float32)
    ref = torch.einsum("eca,ecg,eco,abgo->eb", A, Cw, D, W)
    assert torch.allclose(out, ref, atol=1e-3), (
        (out - ref).abs().max().item())


def test_fold_last_matches_einsum():
    torch.manual_seed(4)
    n, c, P, D = 200, 16, 12, 4
    t = torch.randn(n, c, 3, D, D, device="cuda", requires_grad=True)
    x = torch.randn(n, c, D, device="cuda", requires_grad=True)
    out = fold_last(t, x)
    ref = torch.einsum("ncpqi,nci->ncpq", t, x)
    assert torch.allclose(out, ref, atol=1e-4)
    g = torch.autograd.grad(out.pow(2).sum(), (t, x), create_graph=True)
    gr = torch.autograd.grad(ref.pow(2).sum(), (t, x), create_graph=True)
    for a, b in zip(g, gr):
        assert torch.allclose(a, b, atol=1e-3)


def test_mace_gpu_matches_cpu():
    """Whole MACE forward+forces: GPU fused kernels vs CPU eager."""
    import sys
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from test_mace_model import _build, _mace_config
    from hydragnn_amd.data import Batch
    from hydragnn_amd.ops import scatter
    from hydragnn_amd.utils.datasets.synthetic import md17_shape_dataset

    torch.manual_seed(0)
    dataset = md17_shape_dataset(num_samples=2)
    model, config, _ = _build(_mace_config(), dataset)
    model = model.float()

    def energy_forces(device):
        m = model.to(device)
        batch = Batch.from_data_list(
            [d.clone() for d in dataset]).to(device)
        batch.pos.requires_grad_(True)
        pred = m(batch)
        E = scatter(pred[0], batch.batch, batch.num_graphs, "sum")
        f = -torch.autograd.grad(E.sum(), batch.pos)[0]
        return E.detach().cpu(), f.detach().cpu()

    e_cpu, f_cpu = energy_forces("cpu")
    e_gpu, f_gpu = energy_forces("cuda")
    assert torch.allclose(e_cpu, e_gpu, atol=1e-3, rtol=1e-4), (
        (e_cpu - e_gpu).abs().max())
    assert torch.allclose(f_cpu, f_gpu, atol=1e-3, rtol=1e-3), (
        (f_cpu - f_gpu).abs().max())


def test_etp_fp64_full_precision():
    """fp64 ETP runs on the kernels with double LDS accumulation: match
    the fp64 dense reference to ~1e-12 (not fp32-level error)."""
    torch.manual_seed(0)
    tab = _table()
    E, C = 300, 32
    A = torch.randn(E, C, 4, device="cuda", dtype=torch.float64,
                    requires_grad=True)
    B = torch.randn(E, 9, device="cuda", dtype=torch.float64,
                    requires_grad=True)
    Cw = torch.randn(E, C, 6, device="cuda", dtype=torch.float64)
    out = etp_general(A, B, Cw, tab)
    ref = _dense_general(A.detach(), B.detach(), Cw, tab)
    assert out.dtype == torch.float64
    assert (out - ref).abs().max() < 1e-12
    ga, gb = torch.autograd.grad(out.square().sum(), (A, B),
                                 create_graph=True)
    Ar = A.detach().requires_grad_(True)
    Br = B.detach().requires_grad_(True)
    gar, gbr = torch.autograd.grad(
        _dense_general(Ar, Br, Cw, tab).square().sum(), (Ar, Br),
        create_graph=True)
    assert (ga - gar).abs().max() < 1e-10
    assert (gb - gbr).abs().max() < 1e-10
    gga = torch.autograd.grad(ga.square().sum(), A)[0]
    ggar = torch.autograd.grad(gar.square().sum(), Ar)[0]
    assert (gga - ggar).abs().max() < 1e-9
