"""run_training / run_prediction convenience API (BASELINE north-star
API-compat item) + visualization/dump flags."""

import os

import torch

import hydragnn_amd
from deterministic_graph_data import base_config, make_deterministic_dataset


def test_run_training_and_prediction(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    torch.manual_seed(0)
    cfg = base_config("GIN", heads=("graph",), num_epoch=3)
    cfg["Visualization"] = {"create_plots": True}
    ds = make_deterministic_dataset(num_samples=32, num_heads_node=0)
    monkeypatch.setenv("HYDRAGNN_DUMP_TESTDATA", "1")
    model, out_cfg = hydragnn_amd.run_training(cfg, dataset=ds,
                                               use_gpu=False)
    err, tasks, tv, pv = hydragnn_amd.run_prediction(
        out_cfg, model=model, dataset=ds, use_gpu=False)
    assert float(err) < 1.0
    assert tv[0].shape == pv[0].shape
    # checkpoint + config written under logs/
    logdirs = list((tmp_path / "logs").iterdir())
    assert logdirs, "logs/<name> should exist"
    import glob
    assert glob.glob(str(tmp_path / "logs" / "*" / "*.pk"))
    assert glob.glob(str(tmp_path / "logs" / "*" / "testdata.npz"))
