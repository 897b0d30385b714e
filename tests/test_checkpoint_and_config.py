"""Checkpoint roundtrip, precision control, config normalization
(patterns: reference tests/test_model_loadpred.py:74,
test_precision_control.py:23, test_config.py)."""

import os

import pytest
import torch

from _training_workflow import run_training
from hydragnn_amd.data import Batch
from hydragnn_amd.models import create_model_config
from hydragnn_amd.models.create import resolve_precision
from hydragnn_amd.train import get_head_indices, move_batch_to_device
from hydragnn_amd.utils.config import (
    merge_config,
    update_config,
    update_multibranch_heads,
)
from hydragnn_amd.utils.model import load_existing_model, save_model
from deterministic_graph_data import base_config, make_deterministic_dataset


def test_checkpoint_roundtrip(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    model, config, loaders = run_training(
        "GIN", heads=("graph",), num_samples=24, num_epoch=3)
    opt = torch.optim.AdamW(model.parameters())
    save_model(model, opt, "ckpt_test", path=str(tmp_path / "logs"))
    assert (tmp_path / "logs" / "ckpt_test" / "ckpt_test.pk").exists()

    model2 = create_model_config(config["NeuralNetwork"], use_gpu=False)
    load_existing_model(model2, "ckpt_test", path=str(tmp_path / "logs"))
    batch = next(iter(loaders[2]))
    model.eval(); model2.eval()
    with torch.no_grad():
        p1 = model(batch)[0]
        p2 = model2(batch)[0]
    assert torch.allclose(p1, p2, atol=1e-7), "load/predict mismatch"


def test_checkpoint_epoch_symlink(tmp_path):
    model, config, _ = run_training("GIN", heads=("graph",),
                                    num_samples=16, num_epoch=1)
    opt = torch.optim.AdamW(model.parameters())
    save_model(model, opt, "sym", epoch=3, path=str(tmp_path))
    assert (tmp_path / "sym" / "sym_epoch_3.pk").exists()
    assert (tmp_path / "sym" / "sym.pk").exists()


def test_state_dict_key_structure():
    """Checkpoint-compat: module tree keys follow the reference naming
    (graph_convs.N..., graph_shared.branch-0..., heads_NN.N.branch-0...,
    SURVEY.md hard-part 5)."""
    model, _, _ = run_training("GIN", heads=("graph",), num_samples=8,
                               num_epoch=1)
    keys = list(model.state_dict().keys())
    assert any(k.startswith("graph_convs.0") for k in keys)
    assert any(k.startswith("feature_layers.0") for k in keys)
    assert any("graph_shared.branch-0" in k for k in keys)
    assert any("heads_NN.0.branch-0" in k for k in keys)


@pytest.mark.parametrize("precision,param_dtype,autocast_dtype", [
    ("fp32", torch.float32, None),
    ("bf16", torch.float32, torch.bfloat16),
    ("fp64", torch.float64, None),
])
def test_resolve_precision(precision, param_dtype, autocast_dtype):
    prec, pd, ad = resolve_precision(precision)
    assert pd == param_dtype and ad == autocast_dtype
    torch.set_default_dtype(torch.float32)


def test_fp64_training():
    overrides = {"NeuralNetwork": {"Training": {"precision": "fp64"}}}
    model, config, loaders = run_training(
        "GIN", heads=("graph",), num_samples=16, num_epoch=2,
        overrides=overrides)
    assert next(model.parameters()).dtype == torch.float64
    torch.set_default_dtype(torch.float32)


def test_move_batch_dtype():
    ds = make_deterministic_dataset(num_samples=2, num_heads_node=0)
    b = Batch.from_data_list(ds)
    b = move_batch_to_device(b, torch.float64)
    assert b.x.dtype == torch.float64
    assert b.edge_index.dtype == torch.long  # ints untouched


def test_batches_follow_model_device():
    """Batches go to the MODEL's device, not the globally preferred one
    (a CPU-built model on a GPU box must receive CPU batches)."""
    import torch.nn as nn

    from hydragnn_amd.train.train_validate_test import _model_device

    m = nn.Linear(3, 3)
    assert _model_device(m) == next(m.parameters()).device
    ds = make_deterministic_dataset(num_samples=2, num_heads_node=0)
    b = Batch.from_data_list(ds)
    b = move_batch_to_device(b, torch.float32, _model_device(m))
    assert b.x.device == next(m.parameters()).device
    # no parameters -> falls back to the global device (never raises)
    assert _model_device(nn.Identity()) is not None


def test_update_multibranch_heads():
    out = update_multibranch_heads({"graph": {"num_sharedlayers": 1,
                                              "dim_sharedlayers": 4,
                                              "num_headlayers": 1,
                                              "dim_headlayers": [4]}})
    assert out["graph"][0]["type"] == "branch-0"
    assert "architecture" in out["graph"][0]


def test_merge_config():
    base = {"a": {"b": 1, "c": 2}, "d": 3}
    out = merge_config(base, {"a": {"b": 9}, "e": 4})
    assert out == {"a": {"b": 9, "c": 2}, "d": 3, "e": 4}
    assert base["a"]["b"] == 1  # no mutation


def test_head_indices_multihead():
    from hydragnn_amd.preprocess import create_dataloaders
    config = base_config("GIN", heads=("graph", "node"))
    ds = make_deterministic_dataset(num_samples=4, num_heads_node=1)
    loaders = create_dataloaders(ds, ds, ds, 2, config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"], use_gpu=False)
    batch = next(iter(loaders[0]))
    hidx = get_head_indices(model, batch)
    assert len(hidx) == 2
    total = sum(h.numel() for h in hidx)
    assert total == batch.y.shape[0]
    # indices are disjoint and cover y
    allidx = torch.cat(hidx).sort().values
    assert torch.equal(allidx, torch.arange(batch.y.shape[0]))


def test_equivariant_transformer_config_validation():
    from hydragnn_amd.utils.config import (
        validate_equivariant_transformer_config)
    # SchNet needs scalar-only opt-in
    arch = {"global_attn_engine": "EquivariantTransformer",
            "mpnn_type": "SchNet",
            "equivariant_attn_require_tensor_coupling": True}
    with pytest.raises(ValueError, match="tensor-valued"):
        validate_equivariant_transformer_config(arch)
    arch["equivariant_attn_require_tensor_coupling"] = False
    with pytest.raises(ValueError, match="scalar_only"):
        validate_equivariant_transformer_config(arch)
    arch["equivariant_attn_allow_scalar_only"] = True
    validate_equivariant_transformer_config(arch)  # ok now
    # MACE needs >= 2 layers
    with pytest.raises(ValueError, match="two"):
        validate_equivariant_transformer_config(
            {"global_attn_engine": "EquivariantTransformer",
             "mpnn_type": "MACE", "num_conv_layers": 1})
    # non-ET engine: no-op
    validate_equivariant_transformer_config(
        {"global_attn_engine": "gps", "mpnn_type": "SchNet"})


def test_continue_startfrom(tmp_path, monkeypatch):
    """Training.continue/startfrom restart path (reference
    model.py:204-211)."""
    monkeypatch.chdir(tmp_path)
    torch.manual_seed(0)
    import hydragnn_amd
    cfg = base_config("GIN", heads=("graph",), num_epoch=2)
    ds = make_deterministic_dataset(num_samples=24, num_heads_node=0)
    model, out_cfg = hydragnn_amd.run_training(cfg, dataset=ds,
                                               use_gpu=False)
    from hydragnn_amd.utils.config import get_log_name_config
    name = get_log_name_config(out_cfg)
    cfg2 = base_config("GIN", heads=("graph",), num_epoch=1)
    cfg2["NeuralNetwork"]["Training"]["continue"] = 1
    cfg2["NeuralNetwork"]["Training"]["startfrom"] = name
    # resumed run starts from the checkpointed weights
    model2, _ = hydragnn_amd.run_training(cfg2, dataset=ds,
                                          use_gpu=False)
    assert model2 is not None


def test_evalonly_and_max_batch(monkeypatch, tmp_path):
    monkeypatch.chdir(tmp_path)
    monkeypatch.setenv("HYDRAGNN_EVALONLY", "1")
    model, config, loaders = run_training(
        "GIN", heads=("graph",), num_samples=16, num_epoch=5)
    # eval-only: no training happened -> params at init
    monkeypatch.delenv("HYDRAGNN_EVALONLY")
    monkeypatch.setenv("HYDRAGNN_MAX_NUM_BATCH", "1")
    from hydragnn_amd.train import get_nbatch
    assert get_nbatch(loaders[0]) == 1
    monkeypatch.delenv("HYDRAGNN_MAX_NUM_BATCH")


def test_all_example_configs_parse():
    """Every committed example JSON passes update_config with a tiny
    synthetic dataset (config-schema regression)."""
    import glob
    import json as _json
    import os
    from hydragnn_amd.preprocess import create_dataloaders
    from hydragnn_amd.utils.config import update_config
    from hydragnn_amd.utils.datasets.synthetic import lj_dataset
    import torch
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    configs = sorted(glob.glob(os.path.join(repo, "examples", "*",
                                            "*.json")))
    assert configs, "no example configs found"
    for path in configs:
        with open(path) as f:
            config = _json.load(f)
        if "NeuralNetwork" not in config:
            continue
        ds = lj_dataset(num_samples=8, num_atoms=8, pbc=False)
        # synthesize y/y_loc matching the config's head structure
        types = config["NeuralNetwork"]["Variables_of_interest"].get(
            "type", ["graph"])
        for d in ds:
            spans = [1 if t == "graph" else d.num_nodes for t in types]
            locs = [0]
            for sp in spans:
                locs.append(locs[-1] + sp)
            d.y = torch.zeros(locs[-1], 1)
            d.y_loc = torch.tensor([locs], dtype=torch.long)
        loaders = create_dataloaders(ds, ds, ds, 4, config=config)
        out = update_config(_json.loads(_json.dumps(config)), *loaders)
        assert "output_dim" in out["NeuralNetwork"]["Architecture"], path


def test_checkpoint_best_metric_gating():
    """Checkpoint saves only on improvement after warmup (reference
    model.py:533-573)."""
    from hydragnn_amd.utils.model.model import Checkpoint, EarlyStopping
    ck = Checkpoint("t", warmup=2)
    assert not ck(0, 1.0) and not ck(1, 0.5)   # warmup epochs
    assert ck(2, 0.7)                          # first post-warmup
    assert not ck(3, 0.8)                      # worse -> no save
    assert ck(4, 0.6)                          # better -> save
    es = EarlyStopping(patience=2, min_delta=0.0)
    assert not es(1.0) and not es(1.1)
    assert es(1.2)                             # patience exceeded


def test_update_config_minmax_and_normalize(tmp_path):
    """x/y minmax for output denormalization resolve from the config
    or the serialized container (reference config_utils.py:357-405)."""
    import pickle

    import numpy as np

    from hydragnn_amd.utils.config import (normalize_output_config,
                                           update_config_minmax)

    node_minmax = np.array([[0.0, 1.0, 2.0], [10.0, 11.0, 12.0]])
    graph_minmax = np.array([[-1.0], [5.0]])
    p = tmp_path / "total.pkl"
    with open(p, "wb") as f:
        pickle.dump(node_minmax, f)
        pickle.dump(graph_minmax, f)
        pickle.dump([], f)

    var = {"input_node_features": [0, 2], "type": ["graph", "node"],
           "output_index": [0, 1]}
    out = update_config_minmax(str(p), dict(var))
    assert out["x_minmax"] == [[0.0, 10.0], [2.0, 12.0]]
    assert out["y_minmax"] == [[-1.0, 5.0], [1.0, 11.0]]

    cfg = {"Dataset": {"path": {"total": str(p)}, "name": "t"},
           "NeuralNetwork": {"Variables_of_interest": {
               **var, "denormalize_output": True}}}
    cfg = normalize_output_config(cfg)
    assert cfg["NeuralNetwork"]["Variables_of_interest"]["y_minmax"] \
        == [[-1.0, 5.0], [1.0, 11.0]]

    # minmax provided inline -> no file read
    cfg2 = {"Dataset": {"path": {}, "name": "t"},
            "NeuralNetwork": {"Variables_of_interest": {
                **var, "denormalize_output": True,
                "minmax_node_feature": node_minmax.tolist(),
                "minmax_graph_feature": graph_minmax.tolist()}}}
    cfg2 = normalize_output_config(cfg2)
    assert cfg2["NeuralNetwork"]["Variables_of_interest"]["x_minmax"] \
        == [[0.0, 10.0], [2.0, 12.0]]


def test_check_output_dim_consistent():
    import torch

    from hydragnn_amd.data import Data
    from hydragnn_amd.utils.config import check_output_dim_consistent

    d = Data(x=torch.zeros(4, 1), y=torch.zeros(6, 1),
             y_loc=torch.tensor([[0, 2, 6]]))
    d.num_nodes = 4
    cfg = {"Dataset": {"graph_features": {"dim": [2]},
                       "node_features": {"dim": [0, 1]}},
           "NeuralNetwork": {"Variables_of_interest": {
               "type": ["graph", "node"], "output_index": [0, 1]}}}
    check_output_dim_consistent(d, cfg)  # consistent -> no raise
    cfg["Dataset"]["graph_features"]["dim"] = [3]
    import pytest as _pytest
    with _pytest.raises(AssertionError):
        check_output_dim_consistent(d, cfg)
