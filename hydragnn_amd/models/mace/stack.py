"""MACE stack: O(3)-equivariant higher-order message passing.

Functional parity with /root/reference/hydragnn/models/MACEStack.py:
74-605: per-graph position centering, one-hot(Z,118) node attrs,
spherical-harmonic edge attrs, Bessel x polynomial-cutoff radial
embedding, per-layer interaction (edge tensor product) + product basis
(symmetric contraction) + per-layer multihead readouts summed over
layers; the last layer produces scalars only.
"""

from __future__ import annotations

from typing import Optional

import torch
from torch import nn

from ...ops import (
    get_edge_vectors_and_lengths,
    scatter,
    spherical_harmonics,
)
from ..base import Base
from .blocks import (
    EquivariantProductBasisBlock,
    LinearReadoutBlock,
    NonLinearReadoutBlock,
    RadialEmbeddingBlock,
    RealAgnosticAttResidualInteractionBlock,
    RealAgnosticResidualInteractionBlock,
)
from .o3 import dim

NUM_ELEMENTS = 118


class MACEStack(Base):
    _hipgraph_capture_safe = True  # uses only the given edge_index
    def __init__(
        self,
        r_max: Optional[float] = None,
        radial_type: Optional[str] = "bessel",
        distance_transform: Optional[str] = None,
        num_bessel: Optional[int] = 8,
        max_ell: Optional[int] = 2,
        node_max_ell: Optional[int] = 1,
        avg_num_neighbors: Optional[float] = 10.0,
        envelope_exponent: Optional[int] = 5,
        correlation: Optional[int] = 2,
        edge_dim: Optional[int] = None,
        interaction_type: str = "att",
        **kwargs,
    ):
        self.r_max = r_max or 5.0
        self.radial_type = radial_type or "bessel"
        self.distance_transform = distance_transform
        self.num_bessel = num_bessel or 8
        self.max_ell = max_ell if max_ell is not None else 2
        self.node_max_ell = node_max_ell if node_max_ell is not None else 1
        self.avg_num_neighbors = avg_num_neighbors or 10.0
        self.envelope_exponent = envelope_exponent or 5
        self.correlation = correlation if isinstance(correlation, int) \
            else (correlation[0] if correlation else 2)
        # "att" = reference MACEStack default (blocks.py:121: radial
        # weights attend to endpoint scalars); "residual" = lighter
        # distance-only radial weights
        self.interaction_type = interaction_type or "att"
        self.is_edge_model = True
        super().__init__(edge_dim=edge_dim, **kwargs)

    # ------------------------------------------------------------------
    def _init_conv(self):
        C = self.hidden_dim
        self.node_embedding = nn.Linear(NUM_ELEMENTS, C, bias=False)
        self.radial_embedding = RadialEmbeddingBlock(
            self.r_max, self.num_bessel, self.envelope_exponent,
            self.radial_type, self.distance_transform)
        self.interactions = nn.ModuleList()
        self.products = nn.ModuleList()
        lmax_node = 0
        for ilayer in range(self.num_conv_layers):
            last = ilayer == self.num_conv_layers - 1
            lmax_out = 0 if last else self.node_max_ell
            # interaction always emits the full hidden tower; the
            # product basis contracts to the layer's target (scalars
            # only on the last layer — MACE convention)
            inter_cls = (RealAgnosticAttResidualInteractionBlock
                         if self.interaction_type == "att"
                         else RealAgnosticResidualInteractionBlock)
            self.interactions.append(inter_cls(
                C, lmax_node, self.max_ell, self.node_max_ell,
                self.radial_embedding.out_dim, self.avg_num_neighbors))
            self.products.append(EquivariantProductBasisBlock(
                C, self.node_max_ell, lmax_out,
                self.correlation, NUM_ELEMENTS))
            lmax_node = lmax_out

    def _multihead(self):
        """Per-layer readouts per head, summed over layers
        (reference MACEStack.get_multihead_decoder, :573)."""
        C = self.hidden_dim
        self.graph_shared = nn.ModuleDict({})  # unused; kept for API
        self.num_branches = 1
        self.readouts = nn.ModuleList()
        for ihead in range(self.num_heads):
            per_layer = nn.ModuleList()
            out_dim = self.head_dims[ihead] * (1 + self.var_output)
            for ilayer in range(self.num_conv_layers):
                last = ilayer == self.num_conv_layers - 1
                if last:
                    per_layer.append(NonLinearReadoutBlock(
                        C, max(C // 2, out_dim), out_dim,
                        self.activation_function))
                else:
                    per_layer.append(LinearReadoutBlock(C, out_dim))
            self.readouts.append(per_layer)

    # ------------------------------------------------------------------
    def _node_elements(self, data) -> torch.Tensor:
        """0-based element index from 1-based atomic number Z
        (reference convention Z-1; keeps Z=118 from colliding with 117
        and uses index 0 for hydrogen)."""
        z = data.get("z")
        if z is None:
            z = data.x[:, 0].long()
        return (z.long() - 1).clamp(min=0, max=NUM_ELEMENTS - 1)

    def _embedding(self, data):
        pos = data.pos
        batch = data.get("batch")
        if batch is None:
            batch = torch.zeros(pos.shape[0], dtype=torch.long,
                                device=pos.device)
            data["batch"] = batch
        # center positions per graph (keeps the autograd force path:
        # centering is translation-invariant so forces are unaffected)
        n_graphs = data.get("num_graphs_")
        n_graphs = int(n_graphs) if n_graphs is not None else \
            int(batch.max()) + 1
        mean_pos = scatter(pos, batch, n_graphs, "mean", sorted_index=True)
        pos_c = pos - mean_pos[batch]
        vec, lengths = get_edge_vectors_and_lengths(
            pos_c, data.edge_index, data.get("edge_shifts"))
        edge_sh = spherical_harmonics(vec, self.max_ell, normalize=True)
        elem = self._node_elements(data)
        edge_radial = self.radial_embedding(lengths, z=elem + 1,
                                            edge_index=data.edge_index)
        one_hot = torch.nn.functional.one_hot(
            elem, NUM_ELEMENTS).to(self.node_embedding.weight.dtype)
        h0 = self.node_embedding(one_hot)  # [N, C]
        return h0, elem, edge_sh.to(h0.dtype), edge_radial.to(h0.dtype)

    def _edge_struct(self, data):
        """Per-batch ETP metadata for the fused gather+TP+sum kernel
        (capture-safe index arithmetic, built once per batch)."""
        if not data.pos.is_cuda:
            return None
        key = "_etp_meta_"
        cached = data.get(key)
        if cached is not None:
            return cached
        from ...ops.etp import ETPMeta
        from ...ops.scatter import _rowptr_from_sorted
        src, dst = data.edge_index[0], data.edge_index[1]
        n = data.pos.shape[0]
        eid_d = torch.argsort(dst, stable=True)
        rowptr_dst = _rowptr_from_sorted(dst[eid_d], n)
        meta = ETPMeta(src.numel(), ai=src[eid_d], bi=eid_d, ci=eid_d,
                       rowptr=rowptr_dst, n_a_rows=n)
        # src-sort for gather's backward scatter (src is unsorted in a
        # dst-sorted batch): contention-free indexed CSR instead of
        # atomics
        perm_s = torch.argsort(src, stable=True)
        meta.src_csr = (perm_s, _rowptr_from_sorted(src[perm_s], n))
        meta.dst_csr = (eid_d, rowptr_dst)
        data[key] = meta
        return meta

    def forward(self, data):
        h0, elem, edge_sh, edge_radial = self._embedding(data)
        batch = data["batch"]
        etp_meta = self._edge_struct(data)
        n = h0.shape[0]
        C = self.hidden_dim
        node_feats = h0.view(n, C, 1)
        n_graphs = data.get("num_graphs_")
        n_graphs = int(n_graphs) if n_graphs is not None else \
            int(batch.max()) + 1

        head_outputs = [None] * self.num_heads
        for ilayer, (inter, prod) in enumerate(
                zip(self.interactions, self.products)):
            # pad features to the interaction's input tower
            want = dim(inter.lmax_node)
            if node_feats.shape[-1] < want:
                node_feats = torch.nn.functional.pad(
                    node_feats, (0, want - node_feats.shape[-1]))
            m, sc = inter(node_feats, data.edge_index, edge_sh,
                          edge_radial,
                          edges_sorted=bool(data.get("edges_sorted_",
                                                     False)),
                          etp_meta=etp_meta)
            node_feats = prod(m, elem, sc=sc)
            for ihead in range(self.num_heads):
                r = self.readouts[ihead][ilayer](node_feats)
                head_outputs[ihead] = r if head_outputs[ihead] is None \
                    else head_outputs[ihead] + r

        outputs = []
        outputs_var = []
        for ihead in range(self.num_heads):
            out = head_outputs[ihead]
            hd = self.head_dims[ihead]
            if self.head_type[ihead] == "graph":
                out = self.pool_fn(out, batch, n_graphs)
            outputs.append(out[:, :hd])
            outputs_var.append(out[:, hd:] ** 2 if self.var_output else None)
        if self.var_output:
            return outputs, outputs_var
        return outputs

    def __str__(self):
        return "MACEStack"


def process_node_attributes(node_attributes, num_elements: int):
    """Validate-and-one-hot raw atomic numbers (reference
    MACEStack.process_node_attributes): squeeze to 1-D, warn on
    non-integer or out-of-range values, clamp, one-hot encode."""
    import warnings
    z = node_attributes.squeeze()
    assert z.dim() == 1, (
        "MACE only supports raw atomic numbers as node_attributes")
    if not torch.all(z == z.round()):
        warnings.warn("MACE node_attributes contain non-integer values;"
                      " expected atomic numbers")
    zi = z.round().long()
    if not torch.all((zi >= 1) & (zi <= num_elements)):
        warnings.warn("atomic numbers outside [1, num_elements];"
                      " clamping")
        zi = zi.clamp(1, num_elements)
    return torch.nn.functional.one_hot(zi - 1, num_elements).float()


def get_multihead_decoder(num_channels: int, hidden: int, out_dim: int,
                          nonlinear: bool = True):
    """Reference-named readout factory over this framework's readout
    blocks."""
    from .blocks import LinearReadoutBlock, NonLinearReadoutBlock
    if nonlinear:
        return NonLinearReadoutBlock(num_channels, hidden, out_dim)
    return LinearReadoutBlock(num_channels, out_dim)
