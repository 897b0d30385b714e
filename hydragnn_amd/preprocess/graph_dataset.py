"""Explicit graph-collection pipeline (reference
hydragnn/preprocess/graph_dataset.py): load a pickled list of samples
and prepare each one — radius-graph edges (open or periodic per
sample), predicted-value assembly, atom-feature selection."""

from __future__ import annotations

import pickle

from .graph_samples_checks_and_updates import (
    get_radius_graph, get_radius_graph_pbc, update_atom_features,
    update_predicted_values)


def load_pickled_graphs(dataset_path: str):
    """Load the historical three-object pickle container (minmax node
    features, minmax graph features, sample list)."""
    with open(dataset_path, "rb") as stream:
        pickle.load(stream)
        pickle.load(stream)
        return pickle.load(stream)


def _build_edges(data, radius, max_neighbours, periodic):
    if periodic:
        if getattr(data, "cell", None) is None:
            raise ValueError("Periodic graph samples require data.cell")
        data.pbc = getattr(data, "pbc", (True, True, True))
        return get_radius_graph_pbc(radius, max_neighbours)(data)
    return get_radius_graph(radius, max_neighbours)(data)


def prepare_graph_dataset(dataset, config, dist=False):
    """Per-sample geometry + output assembly per config (reference
    graph_dataset.py:70)."""
    arch = config["NeuralNetwork"]["Architecture"]
    ds_cfg = config["Dataset"]
    variables = config["NeuralNetwork"]["Variables_of_interest"]
    node_features = ds_cfg.get("node_features", {})
    if node_features and not (
            len(node_features.get("name", []))
            == len(node_features.get("dim", []))
            == len(node_features.get("column_index", []))):
        raise ValueError(
            "Node feature names, dimensions, and columns must align")
    radius = arch.get("radius", 5.0)
    max_neighbours = arch.get("max_neighbours", 32)
    periodic = bool(arch.get("periodic_boundary_conditions", False))
    graph_dims = ds_cfg.get("graph_features", {}).get("dim", [])
    node_dims = node_features.get("dim", [])
    for data in dataset:
        _build_edges(data, radius, max_neighbours, periodic)
        if variables.get("type"):
            update_predicted_values(
                variables["type"], variables["output_index"],
                graph_dims, node_dims, data)
        if variables.get("input_node_features") is not None:
            update_atom_features(variables["input_node_features"], data)
    return dataset


def load_and_prepare_graph_dataset(dataset_path, config, dist=False):
    """Load an explicit pickle path and prepare its graph samples."""
    return prepare_graph_dataset(load_pickled_graphs(dataset_path),
                                 config, dist=dist)
