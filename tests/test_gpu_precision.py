"""fp64 and fp16 paths on the GPU: MACE fp64 force parity vs CPU
(ETP kernels dispatch double), and fp16 GradScaler training
(VERDICT r1 items 6/7)."""

import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("requires MI355X GPU", allow_module_level=True)


def _mace_model_and_batch(dtype, n_mols=8):
    from hydragnn_amd.data import Batch
    from hydragnn_amd.utils.datasets.synthetic import md17_shape_dataset
    import bench as B

    torch.manual_seed(2)
    model = B.build_model("cpu", precision="fp32", seed=2).to(dtype)
    ds = md17_shape_dataset(num_samples=n_mols, seed=6)
    batch = Batch.from_data_list([d.clone() for d in ds])
    for k in list(batch.keys()):
        v = batch[k]
        if torch.is_tensor(v) and torch.is_floating_point(v):
            batch[k] = v.to(dtype)
    return model, batch


def _energy_forces(model, batch):
    batch.pos.requires_grad_(True)
    pred = model(batch)
    from hydragnn_amd.ops import scatter
    e = scatter(pred[0].double(), batch.batch, batch.num_graphs,
                "sum").squeeze(-1)
    f = -torch.autograd.grad(e.sum(), batch.pos)[0].double()
    return e.detach(), f.detach()


def test_mace_fp64_gpu_matches_cpu():
    """fp64 MACE on GPU (kernels dispatch double) vs the same model on
    CPU: tight agreement on energies and forces."""
    model, batch = _mace_model_and_batch(torch.float64)
    e_cpu, f_cpu = _energy_forces(model, batch.clone())
    model_g = model.to("cuda")
    batch_g = batch.clone().to("cuda")
    e_gpu, f_gpu = _energy_forces(model_g, batch_g)
    de = (e_cpu - e_gpu.cpu()).abs().max() / e_cpu.abs().max().clamp(min=1)
    df = (f_cpu - f_gpu.cpu()).abs().max() / f_cpu.abs().max().clamp(min=1)
    assert de < 1e-8, f"fp64 energy rel diff {de:.2e}"
    assert df < 1e-6, f"fp64 force rel diff {df:.2e}"


def test_fp16_scaler_training_gpu():
    """fp16 + GradScaler on a graph-head model (the reference's fp16
    usage, /root/reference/hydragnn/train/train_validate_test.py:87-103
    — force double-backward is a bf16/fp32 path): loss decreases and
    the scaler is exercised."""
    from torch.utils.data import DataLoader

    from hydragnn_amd.models.create import create_model
    from hydragnn_amd.preprocess.load_data import _collate
    from hydragnn_amd.train import train
    from hydragnn_amd.train.train_validate_test import (
        get_autocast_and_scaler)
    from hydragnn_amd.utils.datasets.synthetic import (
        md17_shape_dataset_fast)

    autocast, scaler = get_autocast_and_scaler("fp16")
    assert scaler is not None, "fp16 must return a GradScaler"

    torch.manual_seed(3)
    model = create_model(
        mpnn_type="SchNet", input_dim=1, hidden_dim=32,
        output_dim=[1], output_type=["graph"],
        output_heads={"graph": [{"type": "branch-0", "architecture": {
            "num_sharedlayers": 1, "dim_sharedlayers": 32,
            "num_headlayers": 2, "dim_headlayers": [32, 32]}}]},
        activation_function="silu", loss_function_type="mse",
        task_weights=[1.0], num_conv_layers=2, radius=7.0,
        max_neighbours=30, num_gaussians=16, num_filters=32,
        use_gpu=False).to("cuda:0")
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    ds = md17_shape_dataset_fast(64, seed=8, min_dist=0.95,
                                 max_push=40)
    loader = DataLoader(ds, batch_size=16, shuffle=False,
                        collate_fn=_collate)
    errs = []
    for _ in range(4):
        err, _ = train(loader, model, opt, 0, precision="fp16")
        errs.append(float(err))
    assert all(e == e and e < 1e30 for e in errs), errs
    assert errs[-1] < errs[0], f"fp16 loss did not decrease: {errs}"


def test_fused_adamw_kernel_matches_cpu_math():
    """The single-kernel AdamW (fp32 and bf16-master variants) vs the
    CPU flat-math fallback: same trajectory."""
    from hydragnn_amd.ops.fused_adamw import FusedAdamW

    for dtype in (torch.float32, torch.bfloat16):
        torch.manual_seed(7)
        m_gpu = torch.nn.Sequential(
            torch.nn.Linear(8, 32), torch.nn.SiLU(),
            torch.nn.Linear(32, 2)).to("cuda", dtype)
        m_cpu = torch.nn.Sequential(
            torch.nn.Linear(8, 32), torch.nn.SiLU(),
            torch.nn.Linear(32, 2))
        m_cpu.load_state_dict({k: v.float().cpu()
                               for k, v in m_gpu.state_dict().items()})
        m_cpu = m_cpu.to(dtype)
        o_gpu = FusedAdamW(m_gpu.parameters(), lr=2e-3)
        o_cpu = FusedAdamW(m_cpu.parameters(), lr=2e-3)
        x = torch.randn(16, 8).to(dtype)
        y = torch.randn(16, 2).to(dtype)
        for _ in range(5):
            o_gpu.zero_grad()
            torch.nn.functional.mse_loss(
                m_gpu(x.cuda()).float(), y.cuda().float()).backward()
            o_gpu.step()
            o_cpu.zero_grad()
            torch.nn.functional.mse_loss(
                m_cpu(x).float(), y.float()).backward()
            o_cpu.step()
        for pg, pc in zip(m_gpu.parameters(), m_cpu.parameters()):
            tol = 1e-5 if dtype == torch.float32 else 2e-2
            assert torch.allclose(pg.float().cpu(), pc.float(),
                                  rtol=tol, atol=tol), \
                (dtype, (pg.float().cpu() - pc.float()).abs().max())
