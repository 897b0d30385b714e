"""SchNet (continuous-filter conv) stack.

Re-implementation of the SchNet operator (Schütt et al. 2018) with the
reference's capability surface (/root/reference/hydragnn/models/
SCFStack.py:27-332): CFConv with Gaussian smearing + cosine cutoff,
dynamic per-layer radius interaction graph (supports coordinate
updates), optional equivariant coordinate-update branch.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
from torch import nn

from ..ops import (
    cosine_cutoff,
    gather,
    gaussian_basis,
    get_edge_vectors_and_lengths,
    radius_graph,
    scatter,
)
from .base import Base


class ShiftedSoftplus(nn.Module):
    def __init__(self):
        super().__init__()
        self.shift = math.log(2.0)

    def forward(self, x):
        return nn.functional.softplus(x) - self.shift


class GaussianSmearing(nn.Module):
    def __init__(self, start: float, stop: float, num_gaussians: int):
        super().__init__()
        offset = torch.linspace(start, stop, num_gaussians)
        self.coeff = -0.5 / float(offset[1] - offset[0]) ** 2
        self.register_buffer("offset", offset)

    def forward(self, dist):
        return gaussian_basis(dist.view(-1, 1), self.offset.view(1, -1),
                              self.coeff)


class CFConv(nn.Module):
    """x'_i = lin2( sum_j lin1(x_j) * W(e_ij) ), W = filter MLP on
    smeared distances x cosine cutoff."""

    def __init__(self, in_channels, out_channels, num_filters,
                 num_gaussians, cutoff, equivariant_coords: bool = False):
        super().__init__()
        self.cutoff = cutoff
        self.lin1 = nn.Linear(in_channels, num_filters, bias=False)
        self.lin2 = nn.Linear(num_filters, out_channels)
        self.filter_nn = nn.Sequential(
            nn.Linear(num_gaussians, num_filters), ShiftedSoftplus(),
            nn.Linear(num_filters, num_filters))
        self.smearing = GaussianSmearing(0.0, cutoff, num_gaussians)
        self.equivariant_coords = equivariant_coords
        if equivariant_coords:
            layer = nn.Linear(num_filters, 1, bias=False)
            nn.init.xavier_uniform_(layer.weight, gain=0.001)
            self.coord_nn = nn.Sequential(
                nn.Linear(num_filters, num_filters), ShiftedSoftplus(), layer)

    def forward(self, x, pos, edge_index, edge_shifts=None):
        src, dst = edge_index[0], edge_index[1]
        vec, lengths = get_edge_vectors_and_lengths(pos, edge_index,
                                                    edge_shifts)
        d = lengths.squeeze(-1)
        W = self.filter_nn(self.smearing(d).to(x.dtype))
        W = W * cosine_cutoff(d, self.cutoff).view(-1, 1).to(x.dtype)
        msg = gather(self.lin1(x), src) * W
        if self.equivariant_coords:
            trans = vec.to(x.dtype) * self.coord_nn(msg)
            pos = pos + scatter(
                trans, dst, x.shape[0], "mean",
                sorted_index=getattr(self, "_edges_sorted", False)
            ).to(pos.dtype)
        out = scatter(msg, dst, x.shape[0], "sum",
                      sorted_index=getattr(self, "_edges_sorted", False))
        return self.lin2(out), pos


class _SCFWrapper(nn.Module):
    def __init__(self, conv: CFConv, radius: float,
                 max_neighbours: int, dynamic_graph: bool):
        super().__init__()
        self.conv = conv
        self.radius = radius
        self.max_neighbours = max_neighbours
        self.dynamic_graph = dynamic_graph

    def forward(self, inv_node_feat, equiv_node_feat, edge_index,
                edge_attr=None, edge_shifts=None, batch=None, **kwargs):
        if self.dynamic_graph and equiv_node_feat is not None and \
                edge_shifts is None:
            # rebuild the interaction graph from current coordinates
            edge_index = radius_graph(
                equiv_node_feat.detach(), self.radius, batch=batch,
                max_num_neighbors=self.max_neighbours)
        x, pos = self.conv(inv_node_feat, equiv_node_feat, edge_index,
                           edge_shifts)
        return x, pos


class SCFStack(Base):
    def __init__(self, num_gaussians: int = 50, num_filters: int = 64,
                 radius: float = 5.0, max_neighbours: Optional[int] = None,
                 edge_dim: Optional[int] = None, **kwargs):
        self.num_gaussians = num_gaussians or 50
        self.num_filters = num_filters or 64
        self.radius = radius or 5.0
        self.max_neighbours = max_neighbours or 32
        self.is_edge_model = True
        self.scf_equivariance = bool(kwargs.get("equivariance"))
        super().__init__(edge_dim=edge_dim, **kwargs)

    def get_conv(self, input_dim, output_dim, edge_dim=None):
        return _SCFWrapper(
            CFConv(input_dim, output_dim, self.num_filters,
                   self.num_gaussians, self.radius,
                   equivariant_coords=self.scf_equivariance),
            self.radius, self.max_neighbours,
            dynamic_graph=self.scf_equivariance)

    def _embedding(self, data):
        conv_args = {
            "edge_index": data.edge_index,
            "edge_shifts": data.get("edge_shifts"),
            "batch": data.get("batch"),
        }
        x = data.x
        if x is not None and not torch.is_floating_point(x):
            x = x.float()
        if self.use_global_attn and not self.is_equivariant_attn:
            x, conv_args = self._gps_encode(data, x, conv_args)
        return x, data.pos, conv_args

    def __str__(self):
        return "SCFStack"
