"""PaiNN stack: scalar+vector equivariant message passing.

Re-implementation of the PaiNN operator (Schütt et al. 2021) with the
reference's capability surface (/root/reference/hydragnn/models/
PAINNStack.py:27-379): sinc radial basis x cosine cutoff filters, gated
vector messages, scalar-vector update block with <Uv, Vv> coupling,
per-layer output resizing.
"""

from __future__ import annotations

from typing import Optional

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
from .base import Base


class PainnMessage(nn.Module):
    def __init__(self, node_size: int, num_radial: int, cutoff: float):
        super().__init__()
        self.node_size = node_size
        self.cutoff = cutoff
        self.num_radial = num_radial
        self.scalar_message_mlp = nn.Sequential(
            MFMALinear(node_size, node_size), nn.SiLU(),
            MFMALinear(node_size, node_size * 3))
        self.filter_layer = nn.Linear(num_radial, node_size * 3)

    def forward(self, node_scalar, node_vector, edge_index, edge_diff,
                edge_dist):
        src, dst = edge_index[0], edge_index[1]
        rbf = sinc_basis(edge_dist, self.cutoff, self.num_radial)
        filter_weight = self.filter_layer(rbf.to(node_scalar.dtype))
        filter_weight = filter_weight * cosine_cutoff(
            edge_dist, self.cutoff).to(node_scalar.dtype)
        scalar_out = gather(self.scalar_message_mlp(node_scalar), src)
        filter_out = filter_weight * scalar_out
        gate_state_vector, gate_edge_vector, message_scalar = torch.split(
            filter_out, self.node_size, dim=1)

        # vector messages: gated source vectors + gated edge direction
        vj = gather(node_vector.reshape(-1, 3 * self.node_size), src)
        vj = vj.view(-1, 3, self.node_size)
        unit = (edge_diff / edge_dist.clamp(min=1e-9)).to(node_scalar.dtype)
        message_vector = vj * gate_state_vector.unsqueeze(1) + \
            gate_edge_vector.unsqueeze(1) * unit.unsqueeze(-1)

        n = node_scalar.shape[0]
        residual_scalar = scatter(message_scalar, dst, n, "sum",
                                  sorted_index=getattr(self, "_edges_sorted", False))
        residual_vector = scatter(
            message_vector.reshape(-1, 3 * self.node_size), dst, n,
            "sum", sorted_index=getattr(self, "_edges_sorted", False)
        ).view(-1, 3, self.node_size)
        return node_scalar + residual_scalar, node_vector + residual_vector


class PainnUpdate(nn.Module):
    def __init__(self, node_size: int, last_layer: bool = False):
        super().__init__()
        self.node_size = node_size
        self.update_U = nn.Linear(node_size, node_size, bias=False)
        self.update_V = nn.Linear(node_size, node_size, bias=False)
        self.update_mlp = nn.Sequential(
            MFMALinear(node_size * 2, node_size), nn.SiLU(),
            MFMALinear(node_size, node_size * 3))

    def forward(self, node_scalar, node_vector):
        Uv = self.update_U(node_vector)
        Vv = self.update_V(node_vector)
        # eps-safe norm: linalg.norm's backward is v/|v| and NaNs on
        # exactly-zero per-channel vectors (common: vector features
        # start at 0) — fatal for the force double-backward
        Vv_norm = torch.sqrt((Vv * Vv).sum(dim=1) + 1e-12)
        mlp_input = torch.cat([Vv_norm, node_scalar], dim=1)
        mlp_output = self.update_mlp(mlp_input)
        a_vv, a_sv, a_ss = torch.split(mlp_output, self.node_size, dim=1)
        delta_v = a_vv.unsqueeze(1) * Uv
        inner = (Uv * Vv).sum(dim=1)
        delta_s = a_sv * inner + a_ss
        return node_scalar + delta_s, node_vector + delta_v


class _PainnConv(nn.Module):
    def __init__(self, in_size: int, out_size: int, num_radial: int,
                 cutoff: float):
        super().__init__()
        self.in_size = in_size
        self.out_size = out_size
        self.message = PainnMessage(in_size, num_radial, cutoff)
        self.update = PainnUpdate(in_size)
        self.node_embed_out = (nn.Linear(in_size, out_size)
                               if in_size != out_size else nn.Identity())
        self.vec_embed_out = (nn.Linear(in_size, out_size, bias=False)
                              if in_size != out_size else nn.Identity())

    def forward(self, inv_node_feat, equiv_node_feat, edge_index,
                edge_diff, edge_dist, node_vector=None, **kwargs):
        s = inv_node_feat
        if node_vector is None or node_vector.shape[-1] != self.in_size:
            node_vector = s.new_zeros(s.shape[0], 3, self.in_size)
        s, v = self.message(s, node_vector, edge_index, edge_diff, edge_dist)
        s, v = self.update(s, v)
        s = self.node_embed_out(s)
        v = self.vec_embed_out(v)
        return s, v


class _PainnWrapper(nn.Module):
    """Adapter: carries the vector channel inside conv_args state."""

    def __init__(self, conv: _PainnConv):
        super().__init__()
        self.conv = conv

    def forward(self, inv_node_feat, equiv_node_feat, edge_index,
                edge_diff, edge_dist, vec_state, **kwargs):
        s, v = self.conv(inv_node_feat, equiv_node_feat, edge_index,
                         edge_diff, edge_dist,
                         node_vector=vec_state.get("v"))
        vec_state["v"] = v
        return s, equiv_node_feat


class PAINNStack(Base):
    _hipgraph_capture_safe = True  # uses only the given edge_index
    def __init__(self, edge_dim: Optional[int] = None,
                 num_radial: Optional[int] = None,
                 radius: Optional[float] = None, **kwargs):
        self.num_radial = num_radial or 20
        self.radius = radius or 5.0
        self.is_edge_model = True
        # embed raw input into hidden before convs
        self._painn_hidden = kwargs.get("hidden_dim")
        super().__init__(edge_dim=edge_dim, **kwargs)
        self.node_embed = nn.Linear(self.input_dim, self.hidden_dim)

    def _init_conv(self):
        from .base import BatchNormNode
        for _ in range(self.num_conv_layers):
            self.graph_convs.append(self._apply_global_attn(
                _PainnWrapper(_PainnConv(self.hidden_dim, self.hidden_dim,
                                         self.num_radial, self.radius))))
            self.feature_layers.append(BatchNormNode(self.hidden_dim))

    def _embedding(self, data):
        edge_shifts = data.get("edge_shifts")
        edge_diff, edge_dist = get_edge_vectors_and_lengths(
            data.pos, data.edge_index, edge_shifts)
        x = data.x
        if not torch.is_floating_point(x):
            x = x.float()
        x = self.node_embed(x.to(self.node_embed.weight.dtype))
        conv_args = {
            "edge_index": data.edge_index,
            "edge_diff": edge_diff,
            "edge_dist": edge_dist,
            "vec_state": {},
        }
        if self.use_global_attn:
            conv_args["batch"] = data.get("batch")
        return x, data.pos, conv_args

    def __str__(self):
        return "PAINNStack"


def sinc_expansion(edge_dist, edge_size: int, cutoff: float):
    """Reference PAINNStack helper name: sinc radial basis
    sin(n pi d / rc) / d for n = 1..edge_size."""
    return sinc_basis(edge_dist.view(-1, 1), cutoff, edge_size)
