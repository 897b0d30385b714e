"""MLIP wrapper + energy/force consistency (pattern: reference
tests/test_interatomic_potential.py:91-195)."""

import pytest
import torch

from hydragnn_amd.data import Batch
from hydragnn_amd.models import create_model_config
from hydragnn_amd.preprocess import create_dataloaders, split_dataset
from hydragnn_amd.train import train as train_fn
from hydragnn_amd.utils.config import update_config
from hydragnn_amd.utils.datasets.synthetic import lj_dataset, md17_shape_dataset
from hydragnn_amd.utils.optimizer import select_optimizer

from deterministic_graph_data import base_config


def _mlip_config(mpnn_type, num_epoch=6):
    config = base_config(mpnn_type, heads=("node",), num_epoch=num_epoch,
                         hidden_dim=32, lr=0.005, batch_size=8)
    arch = config["NeuralNetwork"]["Architecture"]
    arch["enable_interatomic_potential"] = True
    arch["energy_weight"] = 1.0
    arch["energy_peratom_weight"] = 0.0
    arch["force_weight"] = 10.0
    arch["radius"] = 2.5
    arch["equivariance"] = mpnn_type in ("PAINN", "PNAEq")
    if mpnn_type in ("PAINN", "PNAEq"):
        arch["num_radial"] = 8
    config["NeuralNetwork"]["Variables_of_interest"]["output_dim"] = [1]
    return config


@pytest.mark.parametrize("mpnn_type", ["SchNet", "EGNN", "PAINN",
                                       "PNAEq"])
def test_energy_force_training(mpnn_type):
    # PAINN regression: per-channel zero vectors NaN'd the force
    # double-backward through linalg.norm (now eps-safe)
    torch.manual_seed(3)
    config = _mlip_config(mpnn_type)
    dataset = lj_dataset(num_samples=24, num_atoms=27, pbc=False)
    trainset, valset, testset = split_dataset(dataset, 0.8, seed=0)
    loaders = create_dataloaders(trainset, valset, testset, 8, config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"], use_gpu=False)
    opt = select_optimizer(model,
                           config["NeuralNetwork"]["Training"]["Optimizer"])
    first_err = None
    for epoch in range(6):
        err, tasks = train_fn(loaders[0], model, opt, 0)
        if first_err is None:
            first_err = float(err)
    assert float(err) < first_err, "energy+force loss did not decrease"


def test_force_is_negative_gradient():
    """Forces from the wrapper must equal -dE/dpos of the model."""
    torch.manual_seed(3)
    config = _mlip_config("SchNet")
    dataset = lj_dataset(num_samples=8, num_atoms=27, pbc=False)
    loaders = create_dataloaders(dataset, dataset, dataset, 4, config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"], use_gpu=False)
    batch = Batch.from_data_list(dataset[:4])
    batch.pos.requires_grad_(True)
    pred = model(batch)
    from hydragnn_amd.ops import scatter
    E = scatter(pred[0], batch.batch, batch.num_graphs, "sum").sum()
    f = -torch.autograd.grad(E, batch.pos, create_graph=True)[0]
    # double backward must flow (force training)
    g2 = torch.autograd.grad(f.pow(2).sum(), batch.pos, allow_unused=False)
    assert torch.isfinite(g2[0]).all()


def test_md17_shape_dataset():
    ds = md17_shape_dataset(num_samples=4)
    assert ds[0].num_nodes == 21
    assert ds[0].forces.shape == (21, 3)
    assert torch.isfinite(ds[0].energy).all()


def test_dimenet_energy_force_training():
    """DimeNet MLIP force path (triplet angles under double backward)
    stays finite and optimizes."""
    torch.manual_seed(3)
    config = _mlip_config("DimeNet")
    arch = config["NeuralNetwork"]["Architecture"]
    arch.update({"num_radial": 6, "num_spherical": 7,
                 "basis_emb_size": 8, "int_emb_size": 64,
                 "out_emb_size": 128, "num_before_skip": 1,
                 "num_after_skip": 2})
    dataset = lj_dataset(num_samples=24, num_atoms=27, pbc=False)
    trainset, valset, testset = split_dataset(dataset, 0.8, seed=0)
    loaders = create_dataloaders(trainset, valset, testset, 8,
                                 config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"], use_gpu=False)
    opt = select_optimizer(
        model, config["NeuralNetwork"]["Training"]["Optimizer"])
    errs = []
    for _ in range(4):
        err, _ = train_fn(loaders[0], model, opt, 0)
        errs.append(float(err))
    assert all(e == e for e in errs), errs
