"""End-to-end training integration tests on deterministic synthetic
graphs (pattern + thresholds: reference tests/test_graphs.py:28-313)."""

import pytest
import torch

from _training_workflow import evaluate_error, run_training

# per-model RMSE upper bounds (reference tests/test_graphs.py:147-161)
THRESHOLDS = {
    "GIN": 0.25,
    "SAGE": 0.20,
    "MFC": 0.30,
    "GAT": 0.60,
    "CGCNN": 0.50,
    "PNA": 0.20,
    "PNAPlus": 0.20,
    "SchNet": 0.25,
    "EGNN": 0.20,
    "DimeNet": 0.50,
    "PAINN": 0.60,
    "PNAEq": 0.60,
    "MACE": 0.70,
}

SIMPLE_MODELS = ["GIN", "SAGE", "MFC", "GAT", "CGCNN", "PNA", "PNAPlus",
                 "SchNet", "EGNN", "DimeNet", "PAINN", "PNAEq"]


@pytest.mark.parametrize("mpnn_type", SIMPLE_MODELS)
def test_train_single_graph_head(mpnn_type):
    model, config, loaders = run_training(
        mpnn_type, heads=("graph",), num_samples=64, num_epoch=30)
    err, rmses = evaluate_error(model, loaders[2], config)
    assert rmses[0] < THRESHOLDS[mpnn_type], (
        f"{mpnn_type} graph-head RMSE {rmses[0]:.3f} over threshold")


@pytest.mark.parametrize("mpnn_type", ["GIN", "PNA"])
def test_train_multihead(mpnn_type):
    model, config, loaders = run_training(
        mpnn_type, heads=("graph", "node"), num_samples=64, num_epoch=30)
    err, rmses = evaluate_error(model, loaders[2], config)
    for r in rmses:
        assert r < THRESHOLDS[mpnn_type] + 0.1, (
            f"{mpnn_type} multihead RMSE {rmses} over threshold")
