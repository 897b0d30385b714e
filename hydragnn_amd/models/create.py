"""Model factory + MLIP wrapper.

Behavioral parity with /root/reference/hydragnn/models/create.py:41-819
(create_model_config / create_model / EnhancedModelWrapper with
energy + energy-per-atom + force loss via autograd on positions).
"""

from __future__ import annotations

from typing import List, Optional

import torch

from ..ops import scatter
from ..utils.distributed import get_device
from .stacks import CGCNNStack, GATStack, GINStack, MFCStack, PNAStack, SAGEStack

PRECISION_MAP = {
    "bf16": {"param_dtype": torch.float32, "autocast_dtype": torch.bfloat16},
    # pure bf16: parameters AND activations bf16, no autocast layer —
    # fp32 optimizer master weights via FusedAdamW.  Removes the
    # per-op cast traffic of autocast on memory-bound GNN steps.
    "bf16_pure": {"param_dtype": torch.bfloat16, "autocast_dtype": None},
    "fp16": {"param_dtype": torch.float32, "autocast_dtype": torch.float16},
    "fp32": {"param_dtype": torch.float32, "autocast_dtype": None},
    "fp64": {"param_dtype": torch.float64, "autocast_dtype": None},
}


def resolve_precision(precision):
    if precision is None:
        precision = "fp32"
    prec = str(precision).lower()
    aliases = {"bfloat16": "bf16", "float16": "fp16", "half": "fp16",
               "bf16-pure": "bf16_pure", "pure_bf16": "bf16_pure",
               "float32": "fp32", "float": "fp32",
               "float64": "fp64", "double": "fp64"}
    prec = aliases.get(prec, prec)
    if prec not in PRECISION_MAP:
        raise ValueError(f"Unsupported precision {precision}")
    info = PRECISION_MAP[prec]
    return prec, info["param_dtype"], info["autocast_dtype"]


def create_model(
    mpnn_type: str,
    input_dim: int,
    hidden_dim: int,
    output_dim: List[int],
    output_type: List[str],
    output_heads: dict,
    activation_function: str = "relu",
    loss_function_type: str = "mse",
    task_weights: Optional[List[float]] = None,
    num_conv_layers: int = 2,
    freeze_conv: bool = False,
    initial_bias=None,
    num_nodes: Optional[int] = None,
    max_neighbours: Optional[int] = None,
    edge_dim: Optional[int] = None,
    pna_deg=None,
    pe_dim: int = 0,
    global_attn_engine=None,
    global_attn_type=None,
    global_attn_heads: int = 0,
    num_before_skip=None,
    num_after_skip=None,
    num_radial=None,
    radial_type=None,
    distance_transform=None,
    basis_emb_size=None,
    int_emb_size=None,
    out_emb_size=None,
    envelope_exponent=None,
    num_spherical=None,
    num_gaussians=None,
    num_filters=None,
    radius=None,
    equivariance=None,
    correlation=None,
    max_ell=None,
    node_max_ell=None,
    avg_num_neighbors=None,
    interaction_type=None,
    conv_checkpointing: bool = False,
    enable_interatomic_potential: bool = False,
    energy_weight: float = 0.0,
    energy_peratom_weight: float = 0.0,
    force_weight: float = 0.0,
    use_graph_attr_conditioning: bool = False,
    graph_attr_conditioning_mode: str = "concat_node",
    graph_pooling: str = "mean",
    verbosity: int = 0,
    use_gpu: bool = True,
    **extra,
):
    common = dict(
        input_dim=input_dim,
        hidden_dim=hidden_dim,
        output_dim=output_dim,
        output_type=output_type,
        config_heads=output_heads,
        activation_function_type=activation_function,
        loss_function_type=loss_function_type,
        loss_weights=task_weights,
        num_conv_layers=num_conv_layers,
        freeze_conv=freeze_conv,
        initial_bias=initial_bias,
        num_nodes=num_nodes,
        graph_pooling=graph_pooling,
        pe_dim=pe_dim,
        global_attn_engine=global_attn_engine,
        global_attn_type=global_attn_type,
        global_attn_heads=global_attn_heads,
        use_graph_attr_conditioning=use_graph_attr_conditioning,
        graph_attr_conditioning_mode=graph_attr_conditioning_mode,
    )
    for k in ("equivariant_attn_lmax", "equivariant_attn_num_radial",
              "equivariant_attn_feedforward_multiplier",
              "equivariant_attn_allow_scalar_only",
              "equivariant_attn_require_tensor_coupling",
              "equivariant_attn_chunk_size",
              "equivariant_attn_coupling_mode",
              "graph_attr_dim"):
        if k in extra:
            common[k] = extra[k]

    if mpnn_type == "GIN":
        model = GINStack(equivariance=False, **common)
    elif mpnn_type == "SAGE":
        model = SAGEStack(equivariance=False, **common)
    elif mpnn_type == "MFC":
        model = MFCStack(max_degree=max_neighbours or 10, equivariance=False,
                         **common)
    elif mpnn_type == "GAT":
        model = GATStack(heads=6, negative_slope=0.05, edge_dim=edge_dim,
                         equivariance=False, **common)
    elif mpnn_type == "CGCNN":
        model = CGCNNStack(edge_dim=edge_dim, equivariance=False, **common)
    elif mpnn_type == "PNA":
        assert pna_deg is not None, "PNA requires degree histogram (pna_deg)"
        model = PNAStack(deg=pna_deg, edge_dim=edge_dim, equivariance=False,
                         **common)
    elif mpnn_type == "PNAPlus":
        from .pna_plus import PNAPlusStack
        assert pna_deg is not None
        model = PNAPlusStack(
            deg=pna_deg, edge_dim=edge_dim, envelope_exponent=envelope_exponent,
            num_radial=num_radial, radius=radius, equivariance=False, **common)
    elif mpnn_type == "SchNet":
        from .schnet import SCFStack
        model = SCFStack(
            num_gaussians=num_gaussians, num_filters=num_filters,
            radius=radius, max_neighbours=max_neighbours,
            equivariance=bool(equivariance), edge_dim=edge_dim, **common)
    elif mpnn_type == "EGNN":
        from .egnn import EGCLStack
        model = EGCLStack(edge_dim=edge_dim, max_neighbours=max_neighbours,
                          equivariance=bool(equivariance), **common)
    elif mpnn_type == "DimeNet":
        from .dimenet import DIMEStack
        model = DIMEStack(
            basis_emb_size=basis_emb_size, envelope_exponent=envelope_exponent,
            int_emb_size=int_emb_size, out_emb_size=out_emb_size,
            num_after_skip=num_after_skip, num_before_skip=num_before_skip,
            num_radial=num_radial, num_spherical=num_spherical,
            edge_dim=edge_dim, radius=radius, equivariance=False, **common)
    elif mpnn_type == "PAINN":
        from .painn import PAINNStack
        model = PAINNStack(edge_dim=edge_dim, num_radial=num_radial,
                           radius=radius, equivariance=True, **common)
    elif mpnn_type == "PNAEq":
        from .pnaeq import PNAEqStack
        assert pna_deg is not None
        model = PNAEqStack(deg=pna_deg, edge_dim=edge_dim,
                           num_radial=num_radial, radius=radius,
                           equivariance=True, **common)
    elif mpnn_type == "MACE":
        from .mace import MACEStack
        model = MACEStack(
            r_max=radius, radial_type=radial_type,
            distance_transform=distance_transform,
            num_bessel=num_radial, edge_dim=edge_dim, max_ell=max_ell,
            node_max_ell=node_max_ell, avg_num_neighbors=avg_num_neighbors,
            envelope_exponent=envelope_exponent, correlation=correlation,
            interaction_type=interaction_type or "att",
            equivariance=True, **common)
    else:
        raise ValueError(f"Unknown mpnn_type: {mpnn_type}")

    if conv_checkpointing:
        model.enable_conv_checkpointing()

    if enable_interatomic_potential:
        model = EnhancedModelWrapper(
            model, energy_weight=energy_weight,
            energy_peratom_weight=energy_peratom_weight,
            force_weight=force_weight)

    if use_gpu:
        model = model.to(get_device(use_gpu=use_gpu))
    return model


def create_model_config(config: dict, verbosity: int = 0,
                        use_gpu: bool = True):
    arch = config["Architecture"]
    training = config["Training"]
    model = create_model(
        mpnn_type=arch["mpnn_type"],
        input_dim=arch["input_dim"],
        hidden_dim=arch["hidden_dim"],
        output_dim=arch["output_dim"],
        output_type=arch["output_type"],
        output_heads=arch["output_heads"],
        activation_function=arch.get("activation_function", "relu"),
        loss_function_type=training.get("loss_function_type", "mse"),
        task_weights=arch.get("task_weights"),
        num_conv_layers=arch["num_conv_layers"],
        freeze_conv=arch.get("freeze_conv_layers", False),
        initial_bias=arch.get("initial_bias"),
        num_nodes=arch.get("num_nodes"),
        max_neighbours=arch.get("max_neighbours"),
        edge_dim=arch.get("edge_dim"),
        pna_deg=arch.get("pna_deg"),
        pe_dim=arch.get("pe_dim", 0),
        global_attn_engine=arch.get("global_attn_engine"),
        global_attn_type=arch.get("global_attn_type"),
        global_attn_heads=arch.get("global_attn_heads", 0),
        num_before_skip=arch.get("num_before_skip"),
        num_after_skip=arch.get("num_after_skip"),
        num_radial=arch.get("num_radial"),
        radial_type=arch.get("radial_type"),
        distance_transform=arch.get("distance_transform"),
        basis_emb_size=arch.get("basis_emb_size"),
        int_emb_size=arch.get("int_emb_size"),
        out_emb_size=arch.get("out_emb_size"),
        envelope_exponent=arch.get("envelope_exponent"),
        num_spherical=arch.get("num_spherical"),
        num_gaussians=arch.get("num_gaussians"),
        num_filters=arch.get("num_filters"),
        radius=arch.get("radius"),
        equivariance=arch.get("equivariance"),
        correlation=arch.get("correlation"),
        max_ell=arch.get("max_ell"),
        node_max_ell=arch.get("node_max_ell"),
        avg_num_neighbors=arch.get("avg_num_neighbors"),
        interaction_type=arch.get("interaction_type"),
        conv_checkpointing=training.get("conv_checkpointing", False),
        enable_interatomic_potential=arch.get(
            "enable_interatomic_potential", False),
        energy_weight=arch.get("energy_weight", 0.0),
        energy_peratom_weight=arch.get("energy_peratom_weight", 0.0),
        force_weight=arch.get("force_weight", 0.0),
        use_graph_attr_conditioning=arch.get(
            "use_graph_attr_conditioning", False),
        graph_attr_conditioning_mode=arch.get(
            "graph_attr_conditioning_mode", "concat_node"),
        graph_pooling=arch.get("graph_pooling", "mean"),
        graph_attr_dim=arch.get("graph_attr_dim", 0),
        verbosity=verbosity,
        use_gpu=use_gpu,
        **{k: arch[k] for k in (
            "equivariant_attn_lmax", "equivariant_attn_num_radial",
            "equivariant_attn_feedforward_multiplier",
            "equivariant_attn_allow_scalar_only",
            "equivariant_attn_require_tensor_coupling",
            "equivariant_attn_chunk_size",
            "equivariant_attn_coupling_mode") if k in arch},
    )
    _, param_dtype, _ = resolve_precision(training.get("precision", "fp32"))
    torch.set_default_dtype(param_dtype)
    return model.to(dtype=param_dtype)


def _make_weighted_loss(loss_type: str, w: torch.Tensor):
    """Per-sample weighted mean loss: ``sum(w*err)/sum(w*1)`` with
    ``w`` broadcast from the leading dim.  Equals the unweighted
    nn.MSELoss/L1Loss mean when all weights are 1; zero-weight samples
    contribute neither value nor gradient (used by the static-shape
    pad graph, preprocess/static_batch.py)."""
    lt = (loss_type or "mse").lower()

    def fn(pred, true):
        err = pred.float() - true.float()
        if lt == "mse":
            err = err * err
        elif lt in ("mae", "l1"):
            err = err.abs()
        elif lt in ("smooth_l1", "huber"):
            # beta/delta = 1.0, matching the nn module defaults
            a = err.abs()
            err = torch.where(a < 1.0, 0.5 * err * err, a - 0.5)
        elif lt == "rmse":
            err = err * err  # sqrt applied after the weighted mean
        else:
            raise NotImplementedError(
                f"loss_weight_g not supported for loss '{loss_type}'")
        ww = w.to(err.device)
        while ww.dim() < err.dim():
            ww = ww.unsqueeze(-1)
        ww = ww.expand_as(err)
        out = (err * ww).sum() / ww.sum().clamp_min(1e-12)
        if lt == "rmse":
            out = torch.sqrt(out + 1e-12)
        return out

    return fn


class EnhancedModelWrapper(torch.nn.Module):
    """MLIP composition wrapper: energy + energy/atom + forces
    (forces = -dE/dpos via autograd, create_graph=True during training
    for the double-backward force loss).
    Reference: create.py:671-819."""

    def __init__(self, original_model, energy_weight=0.0,
                 energy_peratom_weight=0.0, force_weight=0.0):
        super().__init__()
        self.model = original_model
        self.energy_weight = energy_weight
        self.energy_peratom_weight = energy_peratom_weight
        self.force_weight = force_weight

    def __getattr__(self, name):
        try:
            return super().__getattr__(name)
        except AttributeError:
            pass
        return getattr(self.model, name)

    def forward(self, data):
        return self.model(data)

    def energy_force_loss(self, pred, data, create_graph=True):
        assert data.get("pos") is not None and data.get("energy") is not None \
            and data.get("forces") is not None, (
                "data.pos, data.energy, data.forces required for "
                "energy-force loss")
        assert data.pos.requires_grad, "data.pos must require grad"
        assert self.num_heads == 1, "Force training requires exactly one head"

        n_graphs = data.get("num_graphs_")
        if n_graphs is None:
            n_graphs = int(data.batch.max()) + 1
        n_graphs = int(n_graphs)
        if self.head_type[0] == "node":
            node_energy_pred = pred[0]
            graph_energy_pred = scatter(
                node_energy_pred, data.batch, n_graphs,
                "sum", sorted_index=True).squeeze(-1).float()
        elif self.head_type[0] == "graph":
            if getattr(self.model, "graph_pooling", "mean") not in ("add",):
                raise ValueError(
                    "Graph head force loss requires sum pooling "
                    "(graph_pooling='add')")
            p = pred[0] if isinstance(pred, (list, tuple)) else pred
            graph_energy_pred = p.squeeze().float()
        else:
            raise ValueError("Force training needs node or graph energy head")

        graph_energy_true = data.energy.squeeze().float()
        # Per-graph loss weights (e.g. the static-shape pad graph at 0,
        # see preprocess/static_batch.py): weighted mean == the
        # unweighted loss over the real graphs, exactly.
        w_g = data.get("loss_weight_g")
        if w_g is not None:
            w_g = w_g.reshape(-1).float()
            loss_fn = _make_weighted_loss(self.loss_function_type, w_g)
        else:
            loss_fn = self.loss_function
        tasks_loss = [loss_fn(graph_energy_pred, graph_energy_true)]

        if (self.energy_weight <= 0 and self.energy_peratom_weight <= 0
                and self.force_weight <= 0):
            raise ValueError(
                "At least one of energy_weight / energy_peratom_weight / "
                "force_weight must be positive")

        tot_loss = 0
        if self.energy_weight > 0:
            tot_loss = tot_loss + tasks_loss[0] * self.energy_weight

        natoms = scatter(torch.ones_like(data.batch,
                                         dtype=graph_energy_pred.dtype),
                         data.batch, n_graphs, "sum",
                         sorted_index=True)
        e_pa_pred = graph_energy_pred / natoms
        e_pa_true = graph_energy_true / natoms
        pa_loss = loss_fn(e_pa_pred, e_pa_true)
        tasks_loss.append(pa_loss)
        if self.energy_peratom_weight > 0:
            tot_loss = tot_loss + pa_loss * self.energy_peratom_weight

        forces_true = data.forces.float()
        forces_pred = torch.autograd.grad(
            graph_energy_pred, data.pos,
            grad_outputs=torch.ones_like(graph_energy_pred),
            retain_graph=graph_energy_pred.requires_grad,
            create_graph=create_graph,
        )[0]
        assert forces_pred is not None
        forces_pred = -forces_pred.float()
        if w_g is not None:
            force_loss_fn = _make_weighted_loss(
                self.loss_function_type, w_g[data.batch])
        else:
            force_loss_fn = loss_fn
        f_loss = force_loss_fn(forces_pred, forces_true)
        tasks_loss.append(f_loss)
        if self.force_weight > 0:
            tot_loss = tot_loss + f_loss * self.force_weight

        return tot_loss, tasks_loss
