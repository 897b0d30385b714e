"""Invariant message-passing stacks: GIN, SAGE, MFC, GAT, CGCNN, PNA.

Capability parity with the reference stacks
(/root/reference/hydragnn/models/{GINStack,SAGEStack,MFCStack,GATStack,
CGCNNStack,PNAStack}.py) on top of our own conv layers (models/layers.py)
and Base skeleton (models/base.py).
"""

from __future__ import annotations

from typing import List

import torch
from torch import nn

from .base import Base, BatchNormNode
from .layers import (
    CGConv,
    GATv2Conv,
    GINConv,
    InvariantConvWrapper,
    MFConv,
    PNAConv,
    SAGEConv,
)


class _StaticShapeStack(Base):
    """Marker base: forward uses only the provided edge_index (no
    dynamic graph rebuild / data-dependent shapes) -> safe to replay
    as a hipGraph across same-shape batches."""

    _hipgraph_capture_safe = True


class GINStack(_StaticShapeStack):
    def get_conv(self, input_dim, output_dim, edge_dim=None):
        mlp = nn.Sequential(
            nn.Linear(input_dim, output_dim),
            nn.ReLU(),
            nn.Linear(output_dim, output_dim),
        )
        return InvariantConvWrapper(GINConv(mlp, eps=100.0, train_eps=True))

    def __str__(self):
        return "GINStack"


class SAGEStack(_StaticShapeStack):
    def get_conv(self, input_dim, output_dim, edge_dim=None):
        return InvariantConvWrapper(SAGEConv(input_dim, output_dim))

    def __str__(self):
        return "SAGEStack"


class MFCStack(_StaticShapeStack):
    def __init__(self, max_degree: int = 10, **kwargs):
        self.max_degree = max_degree
        super().__init__(**kwargs)

    def get_conv(self, input_dim, output_dim, edge_dim=None):
        return InvariantConvWrapper(
            MFConv(input_dim, output_dim, max_degree=self.max_degree))

    def __str__(self):
        return "MFCStack"


class CGCNNStack(_StaticShapeStack):
    """CGConv preserves channel count; hidden_dim is forced equal to
    input_dim by the config normalizer (reference config_utils.py:97-104)."""

    def __init__(self, edge_dim=None, **kwargs):
        self.is_edge_model = True
        super().__init__(edge_dim=edge_dim, **kwargs)

    def get_conv(self, input_dim, output_dim, edge_dim=None):
        return InvariantConvWrapper(
            CGConv(channels=input_dim, dim=edge_dim or 0, aggr="add"),
            uses_edge_attr=self.use_edge_attr)

    def __str__(self):
        return "CGCNNStack"


class PNAStack(_StaticShapeStack):
    def __init__(self, deg: List[int], edge_dim=None, **kwargs):
        self.deg = torch.tensor(deg, dtype=torch.float)
        self.is_edge_model = True
        super().__init__(edge_dim=edge_dim, **kwargs)

    def get_conv(self, input_dim, output_dim, edge_dim=None):
        return InvariantConvWrapper(
            PNAConv(
                input_dim, output_dim,
                aggregators=["mean", "min", "max", "std"],
                scalers=["identity", "amplification", "attenuation", "linear"],
                deg=self.deg, edge_dim=edge_dim,
                pre_layers=1, post_layers=1),
            uses_edge_attr=self.use_edge_attr)

    def __str__(self):
        return "PNAStack"


class GATStack(_StaticShapeStack):
    """GATv2, 6 heads, concat on all but the last conv layer; BatchNorm
    dims are head-aware (reference GATStack.py:39-175)."""

    def __init__(self, heads: int = 6, negative_slope: float = 0.05,
                 edge_dim=None, **kwargs):
        self.heads = heads
        self.negative_slope = negative_slope
        self.is_edge_model = True
        super().__init__(edge_dim=edge_dim, **kwargs)

    def _init_conv(self):
        """All but the last layer concat heads (width = hidden*heads)."""
        self.graph_convs.append(
            self.get_conv(self.embed_dim, self.hidden_dim, concat=True,
                          edge_dim=self.edge_embed_dim))
        self.feature_layers.append(BatchNormNode(self.hidden_dim * self.heads))
        for _ in range(self.num_conv_layers - 2):
            self.graph_convs.append(
                self.get_conv(self.hidden_dim * self.heads, self.hidden_dim,
                              concat=True, edge_dim=self.edge_embed_dim))
            self.feature_layers.append(
                BatchNormNode(self.hidden_dim * self.heads))
        self.graph_convs.append(
            self.get_conv(self.hidden_dim * self.heads, self.hidden_dim,
                          concat=False, edge_dim=self.edge_embed_dim))
        self.feature_layers.append(BatchNormNode(self.hidden_dim))

    def get_conv(self, input_dim, output_dim, concat=False, edge_dim=None):
        return InvariantConvWrapper(
            GATv2Conv(input_dim, output_dim, heads=self.heads, concat=concat,
                      edge_dim=edge_dim,
                      negative_slope=self.negative_slope),
            uses_edge_attr=self.use_edge_attr)

    def __str__(self):
        return "GATStack"
