"""PNAEq stack: PaiNN-style equivariant message passing with PNA
degree-scaler aggregation on the scalar channel.

Reference capability: /root/reference/hydragnn/models/PNAEqStack.py:
41-564 (PainnMessage/PainnUpdate with DegreeScalerAggregation,
safe-sinc radial basis).
"""

from __future__ import annotations

from typing import List, Optional

import torch
from torch import nn

from ..ops import (
    cosine_cutoff,
    gather,
    get_edge_vectors_and_lengths,
    scatter,
    sinc_basis,
)
from ..ops.mfma_linear import MFMALinear
from .base import Base, BatchNormNode
from .layers import DegreeScalerAggregation
from .painn import PainnUpdate


class PNAEqMessage(nn.Module):
    def __init__(self, node_size: int, num_radial: int, cutoff: float,
                 aggregators: List[str], scalers: List[str],
                 deg: torch.Tensor):
        super().__init__()
        self.node_size = node_size
        self.cutoff = cutoff
        self.num_radial = num_radial
        self.scalar_message_mlp = nn.Sequential(
            MFMALinear(node_size, node_size), nn.SiLU(),
            MFMALinear(node_size, node_size * 3))
        self.filter_layer = nn.Linear(num_radial, node_size * 3)
        self.aggr = DegreeScalerAggregation(aggregators, scalers, deg)
        self.scalar_proj = nn.Linear(
            node_size * len(aggregators) * len(scalers), node_size)

    def forward(self, node_scalar, node_vector, edge_index, edge_diff,
                edge_dist):
        src, dst = edge_index[0], edge_index[1]
        rbf = sinc_basis(edge_dist, self.cutoff, self.num_radial)
        fw = self.filter_layer(rbf.to(node_scalar.dtype))
        fw = fw * cosine_cutoff(edge_dist, self.cutoff).to(node_scalar.dtype)
        filter_out = fw * gather(self.scalar_message_mlp(node_scalar), src)
        gate_v, gate_e, msg_s = torch.split(filter_out, self.node_size, dim=1)

        vj = gather(node_vector.reshape(-1, 3 * self.node_size),
                    src).view(-1, 3, self.node_size)
        unit = (edge_diff / edge_dist.clamp(min=1e-9)).to(node_scalar.dtype)
        msg_v = vj * gate_v.unsqueeze(1) + gate_e.unsqueeze(1) * \
            unit.unsqueeze(-1)

        n = node_scalar.shape[0]
        res_s = self.scalar_proj(self.aggr(msg_s, dst, n))
        res_v = scatter(msg_v.reshape(-1, 3 * self.node_size), dst, n,
                        "sum",
                        sorted_index=getattr(self, "_edges_sorted",
                                             False)
                        ).view(-1, 3, self.node_size)
        return node_scalar + res_s, node_vector + res_v


class _PNAEqConv(nn.Module):
    def __init__(self, size: int, num_radial: int, cutoff: float,
                 aggregators, scalers, deg):
        super().__init__()
        self.size = size
        self.message = PNAEqMessage(size, num_radial, cutoff, aggregators,
                                    scalers, deg)
        self.update = PainnUpdate(size)

    def forward(self, inv_node_feat, equiv_node_feat, edge_index,
                edge_diff, edge_dist, vec_state, **kwargs):
        s = inv_node_feat
        v = vec_state.get("v")
        if v is None or v.shape[-1] != self.size:
            v = s.new_zeros(s.shape[0], 3, self.size)
        s, v = self.message(s, v, edge_index, edge_diff, edge_dist)
        s, v = self.update(s, v)
        vec_state["v"] = v
        return s, equiv_node_feat


class PNAEqStack(Base):
    _hipgraph_capture_safe = True  # uses only the given edge_index
    def __init__(self, deg: List[int], edge_dim: Optional[int] = None,
                 num_radial: Optional[int] = None,
                 radius: Optional[float] = None, **kwargs):
        self.deg = torch.tensor(deg, dtype=torch.float)
        self.num_radial = num_radial or 20
        self.radius = radius or 5.0
        self.is_edge_model = True
        super().__init__(edge_dim=edge_dim, **kwargs)
        self.node_embed = nn.Linear(self.input_dim, self.hidden_dim)

    def _init_conv(self):
        aggregators = ["mean", "min", "max", "std"]
        scalers = ["identity", "amplification", "attenuation", "linear"]
        for _ in range(self.num_conv_layers):
            self.graph_convs.append(_PNAEqConv(
                self.hidden_dim, self.num_radial, self.radius, aggregators,
                scalers, self.deg))
            self.feature_layers.append(BatchNormNode(self.hidden_dim))

    def _embedding(self, data):
        edge_diff, edge_dist = get_edge_vectors_and_lengths(
            data.pos, data.edge_index, data.get("edge_shifts"))
        x = data.x
        if not torch.is_floating_point(x):
            x = x.float()
        x = self.node_embed(x.to(self.node_embed.weight.dtype))
        return x, data.pos, {
            "edge_index": data.edge_index,
            "edge_diff": edge_diff,
            "edge_dist": edge_dist,
            "vec_state": {},
        }

    def __str__(self):
        return "PNAEqStack"


class rbf_BasisLayer(torch.nn.Module):
    """Reference PNAEqStack helper name: Bessel RBF x polynomial
    cutoff module."""

    def __init__(self, num_rbf: int, cutoff: float):
        super().__init__()
        import math
        self.cutoff = float(cutoff)
        self.weights = torch.nn.Parameter(
            torch.arange(1, num_rbf + 1).float() * math.pi / cutoff)

    def forward(self, dist):
        from ..ops import bessel_basis, polynomial_cutoff
        r = dist.view(-1, 1)
        return bessel_basis(r, self.cutoff, self.weights) * \
            polynomial_cutoff(r, self.cutoff)
