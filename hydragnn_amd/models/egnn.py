"""E(n)-equivariant GNN stack (EGCL).

Re-implementation of the EGNN operator (Satorras et al. 2021) with the
reference's capability surface (/root/reference/hydragnn/models/
EGCLStack.py:22-300): edge MLP on [h_i, h_j, |r_ij|^2, e_ij], optional
coordinate update coord += mean_j(r_ij * phi_x(m_ij)) when equivariance
is on, node MLP on [h, sum_j m_ij]; PBC edge shifts supported.
"""

from __future__ import annotations

from typing import Optional

import torch
from torch import nn

from ..ops.mfma_linear import MFMALinear

from ..ops import gather, get_edge_vectors_and_lengths, scatter
from .base import Base


class E_GCL(nn.Module):
    def __init__(self, input_nf: int, output_nf: int, hidden_nf: int,
                 edge_dim: int = 0, equivariant: bool = False,
                 act=nn.SiLU()):
        super().__init__()
        self.equivariant = equivariant
        in_edge = 2 * input_nf + 1 + edge_dim
        self.edge_mlp = nn.Sequential(
            MFMALinear(in_edge, hidden_nf), act,
            MFMALinear(hidden_nf, hidden_nf), act)
        self.node_mlp = nn.Sequential(
            MFMALinear(input_nf + hidden_nf, hidden_nf), act,
            MFMALinear(hidden_nf, output_nf))
        if equivariant:
            layer = MFMALinear(hidden_nf, 1, bias=False)
            nn.init.xavier_uniform_(layer.weight, gain=0.001)
            self.coord_mlp = nn.Sequential(
                MFMALinear(hidden_nf, hidden_nf), act, layer)

    def forward(self, h, pos, edge_index, edge_attr=None, edge_shifts=None):
        src, dst = edge_index[0], edge_index[1]
        vec, lengths = get_edge_vectors_and_lengths(pos, edge_index,
                                                    edge_shifts)
        r2 = (lengths ** 2)
        hi = gather(h, dst)
        hj = gather(h, src)
        parts = [hi, hj, r2.to(h.dtype)]
        if edge_attr is not None:
            parts.append(edge_attr)
        m = self.edge_mlp(torch.cat(parts, dim=-1))
        if self.equivariant:
            trans = vec.to(h.dtype) * self.coord_mlp(m)
            pos = pos + scatter(
                trans, dst, h.shape[0], "mean",
                sorted_index=getattr(self, "_edges_sorted", False)
            ).to(pos.dtype)
        agg = scatter(m, dst, h.shape[0], "sum",
                      sorted_index=getattr(self, "_edges_sorted", False))
        h = self.node_mlp(torch.cat([h, agg], dim=-1))
        return h, pos


class _EGCLWrapper(nn.Module):
    def __init__(self, conv: E_GCL):
        super().__init__()
        self.conv = conv

    def forward(self, inv_node_feat, equiv_node_feat, edge_index,
                edge_attr=None, edge_shifts=None, **kwargs):
        h, pos = self.conv(inv_node_feat, equiv_node_feat, edge_index,
                           edge_attr=edge_attr, edge_shifts=edge_shifts)
        return h, pos


class EGCLStack(Base):
    _hipgraph_capture_safe = True  # uses only the given edge_index
    def __init__(self, edge_dim: Optional[int] = None,
                 max_neighbours: Optional[int] = None, **kwargs):
        self.is_edge_model = True
        self.egnn_equivariance = bool(kwargs.get("equivariance"))
        super().__init__(edge_dim=edge_dim, **kwargs)

    def get_conv(self, input_dim, output_dim, edge_dim=None):
        return _EGCLWrapper(E_GCL(
            input_dim, output_dim, output_dim,
            edge_dim=edge_dim or 0,
            equivariant=self.egnn_equivariance))

    def _embedding(self, data):
        conv_args = {
            "edge_index": data.edge_index,
            "edge_shifts": data.get("edge_shifts"),
        }
        if self.use_edge_attr:
            conv_args["edge_attr"] = data.edge_attr
        x = data.x
        if x is not None and not torch.is_floating_point(x):
            x = x.float()
        if self.use_global_attn and not self.is_equivariant_attn:
            x, conv_args = self._gps_encode(data, x, conv_args)
        return x, data.pos, conv_args

    def __str__(self):
        return "EGCLStack"


def unsorted_segment_sum(data, segment_ids, num_segments):
    """Reference EGCLStack helper name; HIP scatter underneath."""
    return scatter(data, segment_ids, num_segments, "sum")
