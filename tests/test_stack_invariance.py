"""Rotation invariance of scalar predictions for the geometric stacks
(pattern: reference per-model equivariant integration tests)."""

import numpy as np
import pytest
import torch

from deterministic_graph_data import base_config, make_deterministic_dataset
from hydragnn_amd.data import Batch
from hydragnn_amd.models import create_model_config
from hydragnn_amd.preprocess import create_dataloaders
from hydragnn_amd.utils.config import update_config


def _rand_rot(seed=0):
    rng = np.random.default_rng(seed)
    Q, _ = np.linalg.qr(rng.normal(size=(3, 3)))
    if np.linalg.det(Q) < 0:
        Q[:, 0] *= -1
    return torch.from_numpy(Q).float()


@pytest.mark.parametrize("mpnn_type", ["PAINN", "PNAEq", "EGNN",
                                       "SchNet", "DimeNet"])
def test_scalar_output_rotation_invariant(mpnn_type):
    torch.manual_seed(0)
    config = base_config(mpnn_type, heads=("graph",), num_epoch=1,
                         hidden_dim=16)
    arch = config["NeuralNetwork"]["Architecture"]
    arch["equivariance"] = mpnn_type in ("PAINN", "PNAEq", "EGNN")
    ds = make_deterministic_dataset(num_samples=8, num_heads_node=0)
    loaders = create_dataloaders(ds, ds, ds, 4, config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"],
                                use_gpu=False).double()
    model.eval()

    batch1 = Batch.from_data_list([d.clone() for d in ds[:4]])
    R = _rand_rot(3).double()
    rot = [d.clone() for d in ds[:4]]
    for d in rot:
        d.pos = d.pos.double() @ R.t()
    batch2 = Batch.from_data_list(rot)
    for b in (batch1, batch2):
        b.pos = b.pos.double()
        b.x = b.x.double()
    with torch.no_grad():
        o1 = model(batch1)[0]
        o2 = model(batch2)[0]
    # note: edge_index identical (distances preserved), so only the
    # geometric features change under rotation
    assert torch.allclose(o1, o2, atol=1e-8), (
        f"{mpnn_type}: {(o1 - o2).abs().max():.2e}")
