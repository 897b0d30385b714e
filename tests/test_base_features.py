"""Base-skeleton feature coverage: conv node heads, mlp_per_node,
vector outputs, graph-attr conditioning (FiLM/concat/fuse), GaussianNLL
variance outputs, loss/activation sweeps, gradient checkpointing
(patterns: reference tests/test_graphs.py variants,
test_loss_and_activation_functions.py, test_graphs_graphattr.py)."""

import pytest
import torch

from _training_workflow import evaluate_error, run_training
from deterministic_graph_data import base_config, make_deterministic_dataset
from hydragnn_amd.models import create_model_config
from hydragnn_amd.preprocess import create_dataloaders
from hydragnn_amd.utils.config import update_config


def _build(config, dataset):
    loaders = create_dataloaders(
        dataset, dataset, dataset,
        config["NeuralNetwork"]["Training"]["batch_size"], config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"], use_gpu=False)
    return model, config, loaders


def test_conv_node_head():
    config = base_config("GIN", heads=("node",), num_epoch=1)
    config["NeuralNetwork"]["Architecture"]["output_heads"]["node"] = {
        "num_headlayers": 2, "dim_headlayers": [8, 8], "type": "conv"}
    ds = make_deterministic_dataset(num_samples=8, num_heads_node=1,
                                    include_graph_head=False)
    model, config, loaders = _build(config, ds)
    batch = next(iter(loaders[0]))
    out = model(batch)
    assert out[0].shape == (batch.num_nodes, 1)
    loss, _ = model.loss(out, batch.y,
                         [torch.arange(batch.y.shape[0])])
    loss.backward()


def test_mlp_per_node_head():
    config = base_config("GIN", heads=("node",), num_epoch=1)
    config["NeuralNetwork"]["Architecture"]["output_heads"]["node"] = {
        "num_headlayers": 1, "dim_headlayers": [8],
        "type": "mlp_per_node"}
    ds = make_deterministic_dataset(num_samples=8, num_heads_node=1,
                                    include_graph_head=False)
    model, config, loaders = _build(config, ds)
    batch = next(iter(loaders[0]))
    out = model(batch)
    assert out[0].shape == (batch.num_nodes, 1)


def test_vector_node_output():
    """3-component node outputs (reference ci_vectoroutput)."""
    config = base_config("EGNN", heads=("node",), num_epoch=1)
    ds = make_deterministic_dataset(num_samples=8, num_heads_node=1,
                                    include_graph_head=False)
    for d in ds:
        n = d.num_nodes
        tgt = d.pos - d.pos.mean(0)
        d.y = tgt.reshape(-1, 1)
        d.y_loc = torch.tensor([[0, 3 * n]])
    model, config, loaders = _build(config, ds)
    assert config["NeuralNetwork"]["Architecture"]["output_dim"] == [3]
    batch = next(iter(loaders[0]))
    out = model(batch)
    assert out[0].shape == (batch.num_nodes, 3)


@pytest.mark.parametrize("mode", ["film", "concat_node", "fuse_pool"])
def test_graph_attr_conditioning(mode):
    config = base_config("GIN", heads=("graph",), num_epoch=2)
    arch = config["NeuralNetwork"]["Architecture"]
    arch["use_graph_attr_conditioning"] = True
    arch["graph_attr_conditioning_mode"] = mode
    arch["graph_attr_dim"] = 2
    ds = make_deterministic_dataset(num_samples=16, num_heads_node=0)
    for d in ds:
        d.graph_attr = torch.rand(1, 2)
    model, config, loaders = _build(config, ds)
    batch = next(iter(loaders[0]))
    out = model(batch)
    # conditioning must actually change the output
    batch2 = next(iter(loaders[0]))
    batch2.graph_attr = batch2.graph_attr + 1.0
    out2 = model(batch2)
    assert not torch.allclose(out[0], out2[0])


def test_gaussian_nll_variance_output():
    config = base_config("GIN", heads=("graph",), num_epoch=1)
    t = config["NeuralNetwork"]["Training"]
    t["loss_function_type"] = "GaussianNLLLoss"
    ds = make_deterministic_dataset(num_samples=8, num_heads_node=0)
    model, config, loaders = _build(config, ds)
    batch = next(iter(loaders[0]))
    out = model(batch)
    assert isinstance(out, tuple) and len(out) == 2
    preds, variances = out
    assert preds[0].shape == variances[0].shape
    assert (variances[0] >= 0).all()
    loss, _ = model.loss(out, batch.y,
                         [torch.arange(batch.y.shape[0])])
    loss.backward()


@pytest.mark.parametrize("loss_fn", ["mse", "mae", "rmse", "huber"])
def test_loss_functions(loss_fn):
    overrides = {"NeuralNetwork": {"Training":
                                   {"loss_function_type": loss_fn}}}
    model, config, loaders = run_training(
        "GIN", heads=("graph",), num_samples=48, num_epoch=20,
        overrides=overrides)
    err, rmses = evaluate_error(model, loaders[2], config)
    assert rmses[0] < 0.4


@pytest.mark.parametrize("act", ["relu", "gelu", "silu", "tanh"])
def test_activation_functions(act):
    overrides = {"NeuralNetwork": {"Architecture":
                                   {"activation_function": act}}}
    model, config, loaders = run_training(
        "GIN", heads=("graph",), num_samples=48, num_epoch=20,
        overrides=overrides)
    err, rmses = evaluate_error(model, loaders[2], config)
    assert rmses[0] < 0.4


def test_conv_checkpointing():
    overrides = {"NeuralNetwork": {"Training":
                                   {"conv_checkpointing": True}}}
    model, config, loaders = run_training(
        "GIN", heads=("graph",), num_samples=48, num_epoch=15,
        overrides=overrides)
    assert model.conv_checkpointing
    err, rmses = evaluate_error(model, loaders[2], config)
    assert rmses[0] < 0.5


@pytest.mark.parametrize("opt", ["SGD", "Adam", "AdamW", "RMSprop"])
def test_optimizers(opt):
    overrides = {"NeuralNetwork": {"Training": {"Optimizer": {
        "type": opt,
        "learning_rate": 0.01 if opt != "SGD" else 0.05}}}}
    model, config, loaders = run_training(
        "GIN", heads=("graph",), num_samples=48, num_epoch=20,
        overrides=overrides)
    err, rmses = evaluate_error(model, loaders[2], config)
    assert rmses[0] < 0.6, f"{opt}: {rmses[0]}"


def test_valtest_and_trace_level_flags(monkeypatch, tmp_path):
    """HYDRAGNN_VALTEST=0 skips val/test epochs; HYDRAGNN_TRACE_LEVEL=1
    adds tracer sync edges without breaking the loop (reference env
    flag surface)."""
    import os
    import sys
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from _training_workflow import run_training
    from hydragnn_amd.utils.profiling_and_tracing import tracer as tr
    monkeypatch.setenv("HYDRAGNN_VALTEST", "0")
    monkeypatch.setenv("HYDRAGNN_TRACE_LEVEL", "1")
    monkeypatch.chdir(tmp_path)
    tr.reset()
    tr.enable()
    model, config, loaders = run_training(
        "GIN", heads=("graph",), num_samples=16, num_epoch=2)
    tr.save(str(tmp_path))
    tr.disable()
    tr.reset()
    assert (tmp_path / "gp_timing.p0").exists()


def test_output_denormalize_and_unscale():
    """postprocess: min-max denormalization per head and per-num-nodes
    unscaling (reference postprocess.py patterns)."""
    import torch
    from hydragnn_amd.data import Data
    from hydragnn_amd.postprocess.postprocess import (
        output_denormalize, unscale_features_by_num_nodes,
        unscale_features_by_num_nodes_config)
    t = [torch.tensor([0.0, 0.5, 1.0])]
    p = [torch.tensor([0.25, 0.5, 0.75])]
    tt, pp = output_denormalize([[10.0, 30.0]], t, p)
    assert torch.allclose(tt[0], torch.tensor([10.0, 20.0, 30.0]))
    assert torch.allclose(pp[0], torch.tensor([15.0, 20.0, 25.0]))
    d = Data(x=torch.ones(4, 1), y=torch.tensor([2.0]),
             pos=torch.zeros(4, 3))
    d.num_nodes = 4
    unscale_features_by_num_nodes(d)
    assert float(d.y) == 8.0
    d2 = Data(x=torch.ones(4, 1), y=torch.tensor([2.0]),
              pos=torch.zeros(4, 3))
    d2.num_nodes = 4
    cfg = {"NeuralNetwork": {"Variables_of_interest": {}}}
    unscale_features_by_num_nodes_config(d2, cfg)  # gate off -> no-op
    assert float(d2.y) == 2.0


def test_training_with_plot_creation(tmp_path, monkeypatch):
    """create_plots=True renders the end-of-training scatter and
    error-histogram files (reference final-plot branch)."""
    pytest.importorskip("matplotlib")
    import os
    import sys
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    monkeypatch.chdir(tmp_path)
    from _training_workflow import run_training
    run_training("GIN", heads=("graph",), num_samples=16, num_epoch=2,
                 train_kwargs={"create_plots": True})
    import glob
    pngs = glob.glob(str(tmp_path / "logs" / "**" / "*.png"),
                     recursive=True)
    assert any("scatter" in p for p in pngs), pngs
    assert any("error_hist" in p for p in pngs), pngs


def test_grad_clip_norm_option():
    """Training.Optimizer.grad_clip_norm clips the global grad norm
    before each step (beyond-reference robustness option)."""
    from hydragnn_amd.utils.optimizer import select_optimizer

    torch.manual_seed(0)
    m = torch.nn.Linear(4, 4)
    opt = select_optimizer(m, {"type": "SGD", "learning_rate": 1.0,
                               "grad_clip_norm": 0.5})
    assert getattr(opt, "_hydragnn_grad_clip", None) == 0.5
    (m(torch.randn(16, 4) * 100).square().sum()).backward()
    import torch as _t
    _t.nn.utils.clip_grad_norm_(m.parameters(), 0.5)
    total = sum(p.grad.norm() ** 2 for p in m.parameters()) ** 0.5
    assert float(total) <= 0.5 + 1e-4


def test_data_from_pyg_ducktype():
    """Data.from_pyg converts any PyG-shaped object (duck-typed keys +
    item access) so reference-era datasets drop into our loaders."""
    import torch

    from hydragnn_amd.data import Batch, Data

    class _PygLike:
        def __init__(self):
            self._d = {"x": torch.randn(5, 2),
                       "edge_index": torch.zeros(2, 3, dtype=torch.long),
                       "y": torch.randn(1, 1)}
            self.num_nodes = 5

        def keys(self):
            return self._d.keys()

        def __getitem__(self, k):
            return self._d[k]

    d = Data.from_pyg(_PygLike())
    assert d.num_nodes == 5 and d.x.shape == (5, 2)
    b = Batch.from_data_list([d, Data.from_pyg(_PygLike())])
    assert b.num_graphs == 2 and b.num_nodes == 10

    d2 = Data.from_dict({"x": torch.ones(3, 1)})
    assert d2.x.shape == (3, 1)
