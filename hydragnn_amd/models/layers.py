"""Message-passing conv layers, built directly on hydragnn_amd.ops
(gather / scatter HIP kernels on GPU).

Re-implementations of the conv semantics the reference pulls from
torch_geometric: GINConv, SAGEConv, MFConv,
GATv2Conv, CGConv, PNAConv (+ degree-scaler aggregation).  Each layer
here follows the published operator definition — not PyG's code.

Uniform stack-facing wrapper signature (see base.Base):
    forward(inv_node_feat, equiv_node_feat, **conv_args) -> (inv, equiv)
"""

from __future__ import annotations

from typing import List, Optional

import torch
from torch import nn

from ..ops import degree, gather, scatter, segment_softmax


class InvariantConvWrapper(nn.Module):
    """Wraps an invariant conv (x, edge_index[, edge_attr]) -> x into the
    uniform (inv, equiv) stack signature."""

    def __init__(self, conv: nn.Module, uses_edge_attr: bool = False):
        super().__init__()
        self.conv = conv
        self.uses_edge_attr = uses_edge_attr

    def forward(self, inv_node_feat, equiv_node_feat, edge_index,
                edge_attr=None, **kwargs):
        if self.uses_edge_attr:
            out = self.conv(inv_node_feat, edge_index, edge_attr)
        else:
            out = self.conv(inv_node_feat, edge_index)
        return out, equiv_node_feat


class GINConv(nn.Module):
    """x'_i = MLP((1 + eps) x_i + sum_j x_j)."""

    def __init__(self, mlp: nn.Module, eps: float = 0.0,
                 train_eps: bool = True):
        super().__init__()
        self.nn = mlp
        if train_eps:
            self.eps = nn.Parameter(torch.tensor(float(eps)))
        else:
            self.register_buffer("eps", torch.tensor(float(eps)))

    def forward(self, x, edge_index):
        src, dst = edge_index[0], edge_index[1]
        agg = scatter(gather(x, src), dst, x.shape[0], "sum",
                      sorted_index=getattr(self, "_edges_sorted", False))
        return self.nn((1.0 + self.eps) * x + agg)


class SAGEConv(nn.Module):
    """x'_i = W1 x_i + W2 mean_j x_j."""

    def __init__(self, in_channels: int, out_channels: int):
        super().__init__()
        self.lin_r = nn.Linear(in_channels, out_channels, bias=True)
        self.lin_l = nn.Linear(in_channels, out_channels, bias=False)

    def forward(self, x, edge_index):
        src, dst = edge_index[0], edge_index[1]
        agg = scatter(gather(x, src), dst, x.shape[0], "mean",
                      sorted_index=getattr(self, "_edges_sorted", False))
        return self.lin_r(x) + self.lin_l(agg)


class MFConv(nn.Module):
    """Molecular fingerprint conv: per-degree weight matrices.
    x'_i = W_{deg(i)} x_i + W'_{deg(i)} sum_j x_j."""

    def __init__(self, in_channels: int, out_channels: int,
                 max_degree: int = 10):
        super().__init__()
        self.max_degree = max_degree
        self.lins_root = nn.ModuleList(
            [nn.Linear(in_channels, out_channels) for _ in range(max_degree + 1)])
        self.lins_nbr = nn.ModuleList(
            [nn.Linear(in_channels, out_channels, bias=False)
             for _ in range(max_degree + 1)])

    def forward(self, x, edge_index):
        src, dst = edge_index[0], edge_index[1]
        agg = scatter(gather(x, src), dst, x.shape[0], "sum",
                      sorted_index=getattr(self, "_edges_sorted", False))
        deg = degree(dst, x.shape[0], torch.long).clamp(max=self.max_degree)
        out = x.new_zeros(x.shape[0], self.lins_root[0].out_features)
        for d in deg.unique().tolist():
            mask = deg == d
            out[mask] = self.lins_root[d](x[mask]) + self.lins_nbr[d](agg[mask])
        return out


class GATv2Conv(nn.Module):
    """GATv2 attention conv (Brody et al. 2022), multi-head."""

    def __init__(self, in_channels: int, out_channels: int, heads: int = 1,
                 concat: bool = True, edge_dim: Optional[int] = None,
                 negative_slope: float = 0.2, dropout: float = 0.0):
        super().__init__()
        self.heads = heads
        self.out_channels = out_channels
        self.concat = concat
        self.negative_slope = negative_slope
        self.dropout = dropout
        self.lin_l = nn.Linear(in_channels, heads * out_channels)
        self.lin_r = nn.Linear(in_channels, heads * out_channels)
        self.att = nn.Parameter(torch.empty(1, heads, out_channels))
        self.edge_lin = (nn.Linear(edge_dim, heads * out_channels)
                         if edge_dim else None)
        bias_dim = heads * out_channels if concat else out_channels
        self.bias = nn.Parameter(torch.zeros(bias_dim))
        nn.init.xavier_uniform_(self.att)

    def forward(self, x, edge_index, edge_attr=None):
        H, C = self.heads, self.out_channels
        src, dst = edge_index[0], edge_index[1]
        xl = self.lin_l(x).view(-1, H, C)   # source transform
        xr = self.lin_r(x).view(-1, H, C)   # target transform
        e = gather(xl.reshape(-1, H * C), src).view(-1, H, C) + \
            gather(xr.reshape(-1, H * C), dst).view(-1, H, C)
        if self.edge_lin is not None and edge_attr is not None:
            e = e + self.edge_lin(edge_attr).view(-1, H, C)
        e_act = torch.nn.functional.leaky_relu(e, self.negative_slope)
        logits = (e_act * self.att).sum(dim=-1)  # [E, H]
        alpha = segment_softmax(logits, dst, x.shape[0])
        if self.training and self.dropout > 0:
            alpha = torch.nn.functional.dropout(alpha, p=self.dropout)
        msg = gather(xl.reshape(-1, H * C), src).view(-1, H, C) * \
            alpha.unsqueeze(-1)
        out = scatter(msg.reshape(-1, H * C), dst, x.shape[0], "sum",
                      sorted_index=getattr(self, "_edges_sorted", False))
        out = out.view(-1, H, C)
        out = out.reshape(-1, H * C) if self.concat else out.mean(dim=1)
        return out + self.bias


class CGConv(nn.Module):
    """Crystal-graph conv (Xie & Grossman 2018):
    x'_i = x_i + sum_j sigma(z W_f) * softplus(z W_s), z=[x_i,x_j,e_ij]."""

    def __init__(self, channels: int, dim: int = 0, aggr: str = "add",
                 bias: bool = True):
        super().__init__()
        self.channels = channels
        self.aggr = aggr
        in_dim = 2 * channels + (dim or 0)
        self.lin_f = nn.Linear(in_dim, channels, bias=bias)
        self.lin_s = nn.Linear(in_dim, channels, bias=bias)

    def forward(self, x, edge_index, edge_attr=None):
        src, dst = edge_index[0], edge_index[1]
        xi = gather(x, dst)
        xj = gather(x, src)
        z = torch.cat([xi, xj] + ([edge_attr] if edge_attr is not None else []),
                      dim=-1)
        msg = torch.sigmoid(self.lin_f(z)) * torch.nn.functional.softplus(
            self.lin_s(z))
        agg = scatter(msg, dst, x.shape[0], self.aggr,
                      sorted_index=getattr(self, "_edges_sorted", False))
        return x + agg


class DegreeScalerAggregation(nn.Module):
    """Multi-aggregator (mean/min/max/std) x degree-scaler
    (identity/amplification/attenuation/linear) concatenation — the PNA
    aggregation (Corso et al. 2020)."""

    def __init__(self, aggregators: List[str], scalers: List[str],
                 deg: torch.Tensor):
        super().__init__()
        self.aggregators = aggregators
        self.scalers = scalers
        deg = deg.to(torch.float)
        num = int(deg.sum())
        bins = torch.arange(deg.numel(), dtype=torch.float)
        self.register_buffer(
            "avg_deg_lin", ((bins * deg).sum() / max(num, 1)).clamp(min=1e-6))
        self.register_buffer(
            "avg_deg_log",
            (((bins + 1).log() * deg).sum() / max(num, 1)).clamp(min=1e-6))

    def forward(self, msg, index, dim_size):
        outs = [scatter(msg, index, dim_size, a,
                        sorted_index=getattr(self, "_edges_sorted",
                                             False))
                for a in self.aggregators]
        out = torch.cat(outs, dim=-1)
        d = degree(index, dim_size, out.dtype).clamp(min=1).view(-1, 1)
        scaled = []
        for s in self.scalers:
            if s == "identity":
                scaled.append(out)
            elif s == "amplification":
                scaled.append(out * ((d + 1).log() / self.avg_deg_log))
            elif s == "attenuation":
                scaled.append(out * (self.avg_deg_log / (d + 1).log()))
            elif s == "linear":
                scaled.append(out * (d / self.avg_deg_lin))
            elif s == "inverse_linear":
                scaled.append(out * (self.avg_deg_lin / d))
            else:
                raise ValueError(f"unknown scaler {s}")
        return torch.cat(scaled, dim=-1)


class PNAConv(nn.Module):
    """Principal Neighbourhood Aggregation conv (single tower,
    divide_input=False — the configuration the reference uses,
    /root/reference/hydragnn/models/PNAStack.py:42-53)."""

    def __init__(self, in_channels: int, out_channels: int,
                 aggregators: List[str], scalers: List[str],
                 deg: torch.Tensor, edge_dim: Optional[int] = None,
                 pre_layers: int = 1, post_layers: int = 1):
        super().__init__()
        self.in_channels = in_channels
        self.edge_dim = edge_dim
        self.aggr = DegreeScalerAggregation(aggregators, scalers, deg)
        pre_in = 3 * in_channels if edge_dim else 2 * in_channels
        mods = [nn.Linear(pre_in, in_channels)]
        for _ in range(pre_layers - 1):
            mods += [nn.ReLU(), nn.Linear(in_channels, in_channels)]
        self.pre_nn = nn.Sequential(*mods)
        if edge_dim:
            self.edge_encoder = nn.Linear(edge_dim, in_channels)
        post_in = in_channels * (1 + len(aggregators) * len(scalers))
        mods = [nn.Linear(post_in, out_channels)]
        for _ in range(post_layers - 1):
            mods += [nn.ReLU(), nn.Linear(out_channels, out_channels)]
        self.post_nn = nn.Sequential(*mods)

    def forward(self, x, edge_index, edge_attr=None):
        src, dst = edge_index[0], edge_index[1]
        xi = gather(x, dst)
        xj = gather(x, src)
        if self.edge_dim and edge_attr is not None:
            z = torch.cat([xi, xj, self.edge_encoder(edge_attr)], dim=-1)
        else:
            z = torch.cat([xi, xj], dim=-1)
        msg = self.pre_nn(z)
        agg = self.aggr(msg, dst, x.shape[0])
        return self.post_nn(torch.cat([x, agg], dim=-1))
