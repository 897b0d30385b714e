"""Every message-passing stack: one forward+backward on the MI355X
(native-kernel path), catching GPU-only device/dtype issues that the
CPU suite cannot."""

import os
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("requires MI355X GPU", allow_module_level=True)

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from _training_workflow import run_training  # noqa: E402
from deterministic_graph_data import base_config, make_deterministic_dataset  # noqa: E402
from hydragnn_amd.data import Batch  # noqa: E402
from hydragnn_amd.models import create_model_config  # noqa: E402
from hydragnn_amd.preprocess import create_dataloaders  # noqa: E402
from hydragnn_amd.train import get_head_indices  # noqa: E402
from hydragnn_amd.utils.config import update_config  # noqa: E402

ALL_STACKS = ["GIN", "SAGE", "MFC", "GAT", "CGCNN", "PNA", "PNAPlus",
              "SchNet", "EGNN", "DimeNet", "PAINN", "PNAEq", "MACE"]


@pytest.mark.parametrize("mpnn_type", ALL_STACKS)
def test_stack_forward_backward_gpu(mpnn_type):
    torch.manual_seed(0)
    config = base_config(mpnn_type, heads=("graph",), num_epoch=1)
    arch = config["NeuralNetwork"]["Architecture"]
    if mpnn_type == "MACE":
        arch.update({"max_ell": 2, "node_max_ell": 1, "correlation": 2,
                     "num_radial": 8})
    ds = make_deterministic_dataset(num_samples=8, num_heads_node=0)
    loaders = create_dataloaders(ds, ds, ds, 4, config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"],
                                use_gpu=False).to("cuda")
    batch = Batch.from_data_list(ds[:4]).to("cuda")
    pred = model(batch)
    head_index = get_head_indices(model, batch)
    loss, _ = model.loss(pred, batch.y, head_index)
    loss.backward()
    for p in model.parameters():
        if p.grad is not None:
            assert torch.isfinite(p.grad).all(), mpnn_type
    assert torch.isfinite(loss), mpnn_type


@pytest.mark.parametrize("engine,mpnn", [("gps", "GIN"),
                                         ("EquivariantTransformer",
                                          "PAINN")])
def test_global_attention_gpu(engine, mpnn):
    torch.manual_seed(0)
    from hydragnn_amd.preprocess import add_laplacian_pe
    config = base_config(mpnn, heads=("graph",), num_epoch=1)
    arch = config["NeuralNetwork"]["Architecture"]
    arch.update({"global_attn_engine": engine,
                 "global_attn_heads": 4, "pe_dim": 3,
                 "equivariant_attn_lmax": 1,
                 "equivariant_attn_num_radial": 8})
    ds = make_deterministic_dataset(num_samples=8, num_heads_node=0)
    if engine == "gps":
        for d in ds:
            add_laplacian_pe(d, 3)
    loaders = create_dataloaders(ds, ds, ds, 4, config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"],
                                use_gpu=False).to("cuda")
    batch = Batch.from_data_list(ds[:4]).to("cuda")
    pred = model(batch)
    loss, _ = model.loss(pred, batch.y, get_head_indices(model, batch))
    loss.backward()
    assert torch.isfinite(loss)


def test_bf16_training_step_gpu():
    """bf16 autocast step on GPU for MACE (the bench precision)."""
    torch.manual_seed(0)
    from test_mace_model import _build, _mace_config
    from hydragnn_amd.utils.datasets.synthetic import md17_shape_dataset
    dataset = md17_shape_dataset(num_samples=8)
    model, config, _ = _build(_mace_config(), dataset)
    model = model.to("cuda")
    batch = Batch.from_data_list(dataset).to("cuda")
    batch.pos.requires_grad_(True)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        pred = model(batch)
        loss, _ = model.energy_force_loss(pred, batch, create_graph=True)
    loss.backward()
    assert torch.isfinite(loss)


@pytest.mark.parametrize("mpnn_type", ALL_STACKS)
def test_stack_gpu_matches_cpu(mpnn_type):
    """HIP-path outputs match the CPU eager reference (fp32)."""
    torch.manual_seed(0)
    config = base_config(mpnn_type, heads=("graph",), num_epoch=1,
                         hidden_dim=16)
    arch = config["NeuralNetwork"]["Architecture"]
    if mpnn_type == "MACE":
        arch.update({"max_ell": 2, "node_max_ell": 1, "correlation": 2,
                     "num_radial": 8})
    ds = make_deterministic_dataset(num_samples=4, num_heads_node=0)
    loaders = create_dataloaders(ds, ds, ds, 4, config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"], use_gpu=False)
    model.eval()
    batch_cpu = Batch.from_data_list([d.clone() for d in ds])
    with torch.no_grad():
        out_cpu = model(batch_cpu)[0]
    model_gpu = model.to("cuda")
    batch_gpu = Batch.from_data_list([d.clone() for d in ds]).to("cuda")
    with torch.no_grad():
        out_gpu = model_gpu(batch_gpu)[0]
    err = (out_cpu - out_gpu.cpu()).abs().max()
    assert err < 1e-3, f"{mpnn_type}: CPU/GPU mismatch {err:.2e}"
