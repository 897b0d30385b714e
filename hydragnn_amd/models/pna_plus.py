"""PNAPlus stack: PNA with Bessel-RBF edge filtering.

Reference capability: /root/reference/hydragnn/models/PNAPlusStack.py:
40-304 — messages are Hadamard-filtered by a projection of a Bessel
radial basis of the edge length before degree-scaler aggregation.
"""

from __future__ import annotations

import math
from typing import List, Optional

import torch
from torch import nn

from ..ops import gather, get_edge_vectors_and_lengths
from .base import Base
from .layers import DegreeScalerAggregation


class BesselBasisLayer(nn.Module):
    def __init__(self, num_radial: int, cutoff: float,
                 envelope_exponent: int = 5):
        super().__init__()
        self.cutoff = cutoff
        self.envelope_exponent = envelope_exponent
        self.freq = nn.Parameter(
            torch.arange(1, num_radial + 1, dtype=torch.float)
            * math.pi / cutoff)

    def envelope(self, x):
        p = self.envelope_exponent + 1
        a = -(p + 1) * (p + 2) / 2
        b = p * (p + 2)
        c = -p * (p + 1) / 2
        xp = x.pow(p)
        return (1.0 / x.clamp(min=1e-9) + a * xp / x.clamp(min=1e-9)
                + b * xp + c * xp * x) * (x < 1.0).to(x.dtype)

    def forward(self, dist):
        d = (dist / self.cutoff).view(-1, 1)
        return self.envelope(d) * torch.sin(self.freq * d)


class PNAPlusConv(nn.Module):
    def __init__(self, in_channels: int, out_channels: int,
                 aggregators: List[str], scalers: List[str],
                 deg: torch.Tensor, num_radial: int,
                 edge_dim: Optional[int] = None):
        super().__init__()
        self.in_channels = in_channels
        self.edge_dim = edge_dim
        self.aggr = DegreeScalerAggregation(aggregators, scalers, deg)
        pre_in = 3 * in_channels if edge_dim else 2 * in_channels
        self.pre_nn = nn.Sequential(nn.Linear(pre_in, in_channels), nn.ReLU(),
                                    nn.Linear(in_channels, in_channels))
        self.rbf_proj = nn.Linear(num_radial, in_channels)
        if edge_dim:
            self.edge_encoder = nn.Linear(edge_dim, in_channels)
        post_in = in_channels * (1 + len(aggregators) * len(scalers))
        self.post_nn = nn.Sequential(
            nn.Linear(post_in, out_channels), nn.ReLU(),
            nn.Linear(out_channels, out_channels))

    def forward(self, x, edge_index, rbf, edge_attr=None):
        src, dst = edge_index[0], edge_index[1]
        xi, xj = gather(x, dst), gather(x, src)
        if self.edge_dim and edge_attr is not None:
            z = torch.cat([xi, xj, self.edge_encoder(edge_attr)], dim=-1)
        else:
            z = torch.cat([xi, xj], dim=-1)
        msg = self.pre_nn(z) * self.rbf_proj(rbf.to(x.dtype))
        agg = self.aggr(msg, dst, x.shape[0])
        return self.post_nn(torch.cat([x, agg], dim=-1))


class _PNAPlusWrapper(nn.Module):
    def __init__(self, conv):
        super().__init__()
        self.conv = conv

    def forward(self, inv_node_feat, equiv_node_feat, edge_index, rbf,
                edge_attr=None, **kwargs):
        return (self.conv(inv_node_feat, edge_index, rbf,
                          edge_attr=edge_attr), equiv_node_feat)


class PNAPlusStack(Base):
    # shape-static forward: uses only the provided edge_index, and the
    # Bessel envelope zeroes edges beyond the cutoff, so padded static
    # batches replay exactly (pad edges are all longer than cutoff)
    _hipgraph_capture_safe = True

    def __init__(self, deg: List[int], edge_dim: Optional[int] = None,
                 envelope_exponent: Optional[int] = None,
                 num_radial: Optional[int] = None,
                 radius: Optional[float] = None, **kwargs):
        self.deg = torch.tensor(deg, dtype=torch.float)
        self.num_radial = num_radial or 5
        self.radius = radius or 5.0
        self.envelope_exponent = envelope_exponent or 5
        self.is_edge_model = True
        super().__init__(edge_dim=edge_dim, **kwargs)
        self.rbf = BesselBasisLayer(self.num_radial, self.radius,
                                    self.envelope_exponent)

    def get_conv(self, input_dim, output_dim, edge_dim=None):
        return _PNAPlusWrapper(PNAPlusConv(
            input_dim, output_dim,
            aggregators=["mean", "min", "max", "std"],
            scalers=["identity", "amplification", "attenuation", "linear"],
            deg=self.deg, num_radial=self.num_radial, edge_dim=edge_dim))

    def _embedding(self, data):
        _, lengths = get_edge_vectors_and_lengths(
            data.pos, data.edge_index, data.get("edge_shifts"))
        rbf = self.rbf(lengths.squeeze(-1))
        conv_args = {"edge_index": data.edge_index, "rbf": rbf}
        if self.use_edge_attr:
            conv_args["edge_attr"] = data.edge_attr
        x = data.x
        if not torch.is_floating_point(x):
            x = x.float()
        return x, data.pos, conv_args

    def __str__(self):
        return "PNAPlusStack"
