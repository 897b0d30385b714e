"""Output denormalization (reference: hydragnn/postprocess/
postprocess.py:13-55)."""

from __future__ import annotations

import torch


def output_denormalize(y_minmax, true_values, predicted_values):
    """Undo min-max normalization per head: v*( max-min ) + min."""
    for ihead in range(len(y_minmax)):
        ymin = torch.tensor(y_minmax[ihead][0])
        ymax = torch.tensor(y_minmax[ihead][1])
        for values in (true_values, predicted_values):
            v = values[ihead]
            values[ihead] = v * (ymax - ymin) + ymin
    return true_values, predicted_values


def unscale_features_by_num_nodes(data):
    """Undo the per-num-nodes scaling applied by the raw-dataset
    pipeline (reference abstractrawdataset.py:305)."""
    n = data.num_nodes
    for key in ("y", "energy"):
        v = data.get(key)
        if v is not None:
            data[key] = v * n
    return data


def unscale_features_by_num_nodes_config(data, config):
    """Config-gated per-node unscaling (reference postprocess.py
    variant): only denormalizes when the config asked for
    total-energy-per-atom style scaling."""
    vo = config["NeuralNetwork"]["Variables_of_interest"]
    if vo.get("denormalize_output", False) or vo.get(
            "scale_features_by_num_nodes", False):
        return unscale_features_by_num_nodes(data)
    return data
