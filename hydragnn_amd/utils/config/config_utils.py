"""JSON-config normalization.

Behavioral parity with the reference config system
(/root/reference/hydragnn/utils/input_config_parsing/config_utils.py:
26-184 update_config, :446 save_config, :474 merge_config) and
update_multibranch_heads (/root/reference/hydragnn/utils/model/
model.py:316): defaults for the ~30 architecture keys, output-dim
inference from y_loc, pna_deg / avg_num_neighbors injection, edge_dim
rules, head-config modernization, deep merge.
"""

from __future__ import annotations

import copy
import json
import os
from typing import Any, Dict, List, Optional

import torch

_ARCH_DEFAULT_NONE = [
    "radius", "radial_type", "distance_transform", "num_gaussians",
    "num_filters", "envelope_exponent", "num_after_skip", "num_before_skip",
    "basis_emb_size", "int_emb_size", "out_emb_size", "num_radial",
    "num_spherical", "correlation", "max_ell", "node_max_ell",
]


def update_multibranch_heads(heads: Dict[str, Any]) -> Dict[str, Any]:
    """Modernize output_heads: dict-style {'graph': {...}} becomes the
    multi-branch list form [{'type': 'branch-0', 'architecture': {...}}]."""
    out = {}
    for level, cfg in heads.items():
        if isinstance(cfg, list):
            branches = []
            for b in cfg:
                if "architecture" in b:
                    branches.append(b)
                else:
                    bid = b.get("type", f"branch-{len(branches)}")
                    branches.append({"type": bid, "architecture": b})
            out[level] = branches
        else:
            out[level] = [{"type": "branch-0", "architecture": cfg}]
    return out


def check_if_graph_size_variable(*loaders) -> bool:
    sizes = set()
    for loader in loaders:
        if loader is None:
            continue
        ds = loader.dataset
        n = min(len(ds), 50)
        for i in range(n):
            sizes.add(ds[i].num_nodes)
        if len(sizes) > 1:
            return True
    return len(sizes) > 1


def _gather_deg(dataset) -> torch.Tensor:
    from ...ops import degree
    max_deg = 0
    hists = []
    for data in dataset:
        d = degree(data.edge_index[1], data.num_nodes, torch.long)
        hists.append(torch.bincount(d))
        max_deg = max(max_deg, hists[-1].numel())
    out = torch.zeros(max_deg, dtype=torch.long)
    for h in hists:
        out[: h.numel()] += h
    if torch.distributed.is_initialized():
        from ..distributed import to_comm_device
        t, moved = to_comm_device(out)
        torch.distributed.all_reduce(t)
        out = t.cpu() if moved else t
    return out


def _calculate_avg_deg(dataset) -> float:
    num_edges = 0
    num_nodes = 0
    for data in dataset:
        num_edges += data.num_edges
        num_nodes += data.num_nodes
    t = torch.tensor([num_edges, num_nodes], dtype=torch.float64)
    if torch.distributed.is_initialized():
        from ..distributed import to_comm_device
        t, _ = to_comm_device(t)
        torch.distributed.all_reduce(t)
    return float(t[0] / t[1].clamp(min=1))


def update_config(config, train_loader, val_loader, test_loader):
    """Normalize the raw JSON config with dataset-derived fields."""
    gsv = os.getenv("HYDRAGNN_USE_VARIABLE_GRAPH_SIZE")
    if gsv is None:
        graph_size_variable = check_if_graph_size_variable(
            train_loader, val_loader, test_loader)
    else:
        graph_size_variable = bool(int(gsv))

    nn = config["NeuralNetwork"]
    arch = nn["Architecture"]
    training = nn["Training"]
    var = nn["Variables_of_interest"]

    arch.setdefault("global_attn_engine", None)
    arch.setdefault("global_attn_type", None)
    arch.setdefault("global_attn_heads", 0)
    arch.setdefault("pe_dim", 0)
    training.setdefault("global_attn_redraw_interval", 1000)
    arch.setdefault("equivariant_attn_lmax", 1)
    arch.setdefault("equivariant_attn_num_radial", 16)
    arch.setdefault("equivariant_attn_feedforward_multiplier", 2)
    arch.setdefault("equivariant_attn_allow_scalar_only", False)
    arch.setdefault("equivariant_attn_require_tensor_coupling", True)
    arch.setdefault("equivariant_attn_chunk_size", 512)
    arch.setdefault("equivariant_attn_coupling_mode", "parallel")

    validate_equivariant_transformer_config(arch)

    batching = training.get("Batching")
    if batching is not None:
        mode = batching.get("mode", "fixed")
        if mode not in ("fixed", "node_budget"):
            raise ValueError(f"unsupported batching mode: {mode}")
        if mode == "node_budget" and "max_nodes" not in batching:
            raise ValueError("node_budget batching requires max_nodes")

    arch["output_heads"] = update_multibranch_heads(arch["output_heads"])

    # --- output dims from data / y_loc ---
    data0 = train_loader.dataset[0]
    update_config_NN_outputs(config, data0, graph_size_variable)

    var.setdefault("denormalize_output", False)

    arch["input_dim"] = len(var["input_node_features"])

    # --- PNA degree histogram / MACE avg_num_neighbors ---
    if arch["mpnn_type"] in ("PNA", "PNAPlus", "PNAEq"):
        if hasattr(train_loader.dataset, "pna_deg") and \
                train_loader.dataset.pna_deg is not None:
            deg = torch.tensor(train_loader.dataset.pna_deg)
        else:
            deg = _gather_deg(train_loader.dataset)
        arch["pna_deg"] = deg.tolist()
        arch["max_neighbours"] = len(deg) - 1
    else:
        arch["pna_deg"] = None

    if arch["mpnn_type"] == "CGCNN" and not arch.get("global_attn_engine"):
        arch["hidden_dim"] = arch["input_dim"]

    if arch["mpnn_type"] == "MACE":
        if hasattr(train_loader.dataset, "avg_num_neighbors") and \
                train_loader.dataset.avg_num_neighbors is not None:
            arch["avg_num_neighbors"] = float(
                train_loader.dataset.avg_num_neighbors)
        else:
            arch["avg_num_neighbors"] = _calculate_avg_deg(
                train_loader.dataset)
    else:
        arch["avg_num_neighbors"] = None

    for key in _ARCH_DEFAULT_NONE:
        arch.setdefault(key, None)
    arch.setdefault("enable_interatomic_potential", False)

    # --- edge dim rules ---
    update_config_edge_dim(config)

    arch.setdefault("equivariance", None)
    arch.setdefault("freeze_conv_layers", False)
    arch.setdefault("initial_bias", None)
    arch.setdefault("activation_function", "relu")
    arch.setdefault("SyncBatchNorm", False)
    training.setdefault("conv_checkpointing", False)
    training.setdefault("loss_function_type", "mse")
    training.setdefault("Optimizer", {"type": "AdamW", "learning_rate": 1e-3})
    training.setdefault("precision", "fp32")
    return config


def update_config_NN_outputs(config, data, graph_size_variable):
    """Derive per-head output dims from the sample's ``y_loc`` packing
    (reference config_utils.py:313-355)."""
    nn = config["NeuralNetwork"]
    arch = nn["Architecture"]
    var = nn["Variables_of_interest"]
    output_type = var["type"]
    if arch.get("enable_interatomic_potential", False):
        dims_list = var["output_dim"]
    elif data.get("y_loc") is not None:
        dims_list = []
        for ihead in range(len(output_type)):
            span = int(data.y_loc[0, ihead + 1]) - int(data.y_loc[0, ihead])
            if output_type[ihead] == "graph":
                dims_list.append(span)
            elif output_type[ihead] == "node":
                if (graph_size_variable and
                        arch["output_heads"]["node"][0]["architecture"]["type"]
                        == "mlp_per_node"):
                    raise ValueError(
                        "mlp_per_node not allowed for variable graph size")
                dims_list.append(span // data.num_nodes)
            else:
                raise ValueError(f"Unknown output type {output_type[ihead]}")
    else:
        for t in output_type:
            if t != "graph":
                raise ValueError("y_loc needed for non-graph outputs")
        dims_list = var["output_dim"]
    arch["output_dim"] = dims_list
    arch["output_type"] = output_type
    arch["num_nodes"] = data.num_nodes
    return config


def update_config_edge_dim(config):
    """Edge-feature dimension rules per architecture (reference
    config_utils.py:265-292)."""
    arch = config["NeuralNetwork"]["Architecture"]
    arch["edge_dim"] = None
    edge_models = ["GAT", "PNA", "PNAPlus", "PAINN", "PNAEq", "CGCNN",
                   "SchNet", "EGNN", "DimeNet", "MACE"]
    if arch.get("edge_features"):
        assert arch["mpnn_type"] in edge_models, (
            "Edge features only with " + ",".join(edge_models))
        arch["edge_dim"] = len(arch["edge_features"])
        assert not arch.get("enable_interatomic_potential", False), (
            "Edge features cannot be combined with interatomic potentials")
    elif arch["mpnn_type"] == "CGCNN":
        arch["edge_dim"] = 0
    return config


def update_config_equivariance(arch):
    """Equivariance flag normalization (reference config_utils.py:252):
    only EGNN toggles behavior on it; other architectures are either
    inherently equivariant (PaiNN/PNAEq/MACE) or ignore the flag, so a
    stray setting warns instead of raising."""
    import warnings
    toggled = ["EGNN"]
    inherently = ["SchNet", "PAINN", "PNAEq", "MACE"]
    if "equivariance" in arch:
        if arch.get("equivariance") and arch["mpnn_type"] not in (
                toggled + inherently):
            warnings.warn(
                f"E(3) equivariance is only toggled for {toggled}; "
                f"setting it for {arch['mpnn_type']} has no effect")
    else:
        arch["equivariance"] = None
    return arch


def check_output_dim_consistent(data, config):
    """Assert the sample's y_loc spans match the Dataset feature dims
    named by output_index (reference config_utils.py:295-310)."""
    var = config["NeuralNetwork"]["Variables_of_interest"]
    output_type = var["type"]
    output_index = var["output_index"]
    if data.get("y_loc") is None:
        return
    for ihead in range(len(output_type)):
        span = int(data.y_loc[0, ihead + 1]) - int(data.y_loc[0, ihead])
        if output_type[ihead] == "graph":
            expect = config["Dataset"]["graph_features"]["dim"][
                output_index[ihead]]
            assert span == expect, (ihead, span, expect)
        elif output_type[ihead] == "node":
            expect = config["Dataset"]["node_features"]["dim"][
                output_index[ihead]]
            assert span // data.num_nodes == expect, (ihead, span, expect)


def update_config_minmax(dataset_path, var_config):
    """Populate ``x_minmax``/``y_minmax`` for output denormalization
    (reference config_utils.py:381-405): minmax arrays come from the
    config when present, else from the serialized dataset container's
    leading two pickle objects (minmax node features, minmax graph
    features)."""
    import pickle

    import numpy as np
    if var_config.get("minmax_node_feature") is not None and \
            var_config.get("minmax_graph_feature") is not None:
        node_minmax = np.asarray(var_config["minmax_node_feature"])
        graph_minmax = np.asarray(var_config["minmax_graph_feature"])
    else:
        with open(dataset_path, "rb") as f:
            node_minmax = np.asarray(pickle.load(f))
            graph_minmax = np.asarray(pickle.load(f))
    var_config["x_minmax"] = [
        node_minmax[:, i].tolist()
        for i in var_config["input_node_features"]]
    var_config["y_minmax"] = []
    for out_t, out_i in zip(var_config["type"],
                            var_config["output_index"]):
        if out_t == "graph":
            var_config["y_minmax"].append(graph_minmax[:, out_i].tolist())
        elif out_t == "node":
            var_config["y_minmax"].append(node_minmax[:, out_i].tolist())
        else:
            raise ValueError(f"Unknown output type {out_t}")
    return var_config


def normalize_output_config(config):
    """Resolve the denormalization minmax source from the Dataset
    paths and fill Variables_of_interest (reference
    config_utils.py:357-378)."""
    var = config["NeuralNetwork"]["Variables_of_interest"]
    if var.get("denormalize_output"):
        if var.get("minmax_node_feature") is not None and \
                var.get("minmax_graph_feature") is not None:
            dataset_path = None
        else:
            paths = config["Dataset"]["path"]
            first = str(list(paths.values())[0])
            if first.endswith(".pkl"):
                dataset_path = first
            else:
                base = os.environ.get("SERIALIZED_DATA_PATH", os.getcwd())
                name = config["Dataset"]["name"]
                suffix = "" if "total" in paths else "_train"
                dataset_path = (f"{base}/serialized_dataset/"
                                f"{name}{suffix}.pkl")
        config["NeuralNetwork"]["Variables_of_interest"] = \
            update_config_minmax(dataset_path, var)
    else:
        var["denormalize_output"] = False
    return config


def validate_equivariant_transformer_config(arch) -> None:
    """Engine-specific option validation (reference
    config_utils.py:187-250)."""
    if arch.get("global_attn_engine") != "EquivariantTransformer":
        return
    mpnn_type = arch.get("mpnn_type")
    if mpnn_type in ("SchNet", "DimeNet"):
        if arch.get("equivariant_attn_require_tensor_coupling", True):
            raise ValueError(
                f"{mpnn_type} cannot provide tensor-valued local/global "
                "coupling; set equivariant_attn_require_tensor_coupling="
                "false")
        if not arch.get("equivariant_attn_allow_scalar_only", False):
            raise ValueError(
                f"{mpnn_type} requires "
                "equivariant_attn_allow_scalar_only=true")
    if mpnn_type == "SchNet" and arch.get("equivariance"):
        raise ValueError(
            "SchNet with EquivariantTransformer cannot use coordinate "
            "updates; set Architecture.equivariance=false")
    if mpnn_type == "MACE" and arch.get("num_conv_layers", 0) < 2:
        raise ValueError(
            "MACE with EquivariantTransformer requires at least two "
            "convolution layers (the final MACE layer is scalar-only)")
    lmax = arch.get("equivariant_attn_lmax", 1)
    if not (0 <= int(lmax) <= 3):
        raise ValueError("equivariant_attn_lmax must be in [0, 3]")


def save_config(config, log_name: str, path: str = "./logs/") -> None:
    fname = os.path.join(path, log_name, "config.json")
    os.makedirs(os.path.dirname(fname), exist_ok=True)
    rank = 0
    if torch.distributed.is_initialized():
        rank = torch.distributed.get_rank()
    if rank == 0:
        with open(fname, "w") as f:
            json.dump(config, f, indent=2)


def merge_config(base: Dict, override: Dict) -> Dict:
    """Deep merge (override wins)."""
    out = copy.deepcopy(base)
    for k, v in override.items():
        if k in out and isinstance(out[k], dict) and isinstance(v, dict):
            out[k] = merge_config(out[k], v)
        else:
            out[k] = copy.deepcopy(v)
    return out


def get_log_name_config(config) -> str:
    arch = config["NeuralNetwork"]["Architecture"]
    training = config["NeuralNetwork"]["Training"]
    name = config.get("Dataset", {}).get("name", "dataset")
    cut = name.rfind("_") if name.rfind("_") > 0 else None
    return (
        f"{arch['mpnn_type']}-r-{arch.get('radius')}"
        f"-ncl-{arch['num_conv_layers']}-hd-{arch['hidden_dim']}"
        f"-ne-{training['num_epoch']}"
        f"-lr-{training['Optimizer']['learning_rate']}"
        f"-bs-{training['batch_size']}"
        f"-data-{name[:cut]}"
    )


def parse_deepspeed_config(config) -> Dict:
    """DeepSpeed engine config (reference config_utils.py:455-471):
    a user-supplied ``NeuralNetwork.ds_config`` section passes
    through; otherwise the per-GPU micro batch defaults to
    ``Training.batch_size`` with no gradient accumulation.  The
    PER-GPU key matters: a global ``train_batch_size`` equal to the
    local batch fails DeepSpeed's micro*accum*world consistency check
    at world_size > 1.  No ``optimizer`` section — the wrapper passes
    the already-built optimizer instance to ``deepspeed.initialize``
    and DeepSpeed rejects specifying both.  On MI355X the native path
    is DDP/FSDP over RCCL."""
    ds_config = dict(config["NeuralNetwork"].get("ds_config", {}))
    if "train_micro_batch_size_per_gpu" not in ds_config and \
            "train_batch_size" not in ds_config:
        training = config["NeuralNetwork"]["Training"]
        ds_config["train_micro_batch_size_per_gpu"] = \
            training.get("batch_size", 32)
        ds_config["gradient_accumulation_steps"] = 1
    if "steps_per_print" not in ds_config:
        ds_config["steps_per_print"] = int(1e9)  # disable printing
    return ds_config
