"""Base multi-headed GNN skeleton.

MI355X-native re-design of the reference Base model
(/root/reference/hydragnn/models/Base.py:37-1048): embedding -> N conv
layers (BatchNorm + activation, optional gradient checkpointing) ->
graph pooling -> per-head decoders (graph MLP heads through shared dense
layers, node MLP/conv heads), multi-branch (per-dataset) masking,
graph_attr conditioning (FiLM / concat_node / fuse_pool), weighted
multi-task loss, optional GaussianNLL variance outputs.

Differences from the reference by design:
  - conv layers have the uniform signature
    conv(inv_node_feat, equiv_node_feat, **conv_args) -> (inv, equiv)
    instead of PyG string-Sequential plumbing;
  - all aggregation funnels through hydragnn_amd.ops (HIP kernels on
    GPU);
  - module attribute names (graph_convs.N, feature_layers.N,
    graph_shared.branch-0, heads_NN.N.branch-0) match the reference so
    checkpoint state dicts keep the same key structure.
"""

from __future__ import annotations

from typing import List, Optional

import torch
from torch import nn
from torch.nn import Module, ModuleDict, ModuleList, Sequential

# Drop-in nn.Linear that dispatches to the MFMA / narrow-output GEMV
# HIP kernels when shapes qualify (falls back to split-K / F.linear):
# keeps decoder-head GEMMs off hipBLASLt's degenerate N=1 tiles.
from ..ops.mfma_linear import MFMALinear as Linear
from torch.utils.checkpoint import checkpoint

from ..ops import scatter
from ..utils.model.activation import (
    activation_function_selection,
    loss_function_selection,
)


class BatchNormNode(nn.BatchNorm1d):
    """BatchNorm over the node dimension (PyG BatchNorm equivalent)."""


def global_mean_pool(x, batch, size=None):
    return scatter(x, batch, size, "mean", sorted_index=True)


def global_add_pool(x, batch, size=None):
    # the batch vector is sorted by construction -> CSR segment kernel
    return scatter(x, batch, size, "sum", sorted_index=True)


def global_max_pool(x, batch, size=None):
    return scatter(x, batch, size, "max")


class Base(Module):
    # stacks whose forward is shape-static given a fixed batch shape
    # override this to True (see train/captured.py)
    _hipgraph_capture_safe = False

    def __init__(
        self,
        input_dim: int,
        hidden_dim: int,
        output_dim: List[int],
        output_type: List[str],
        config_heads: dict,
        activation_function_type: str = "relu",
        loss_function_type: str = "mse",
        equivariance: bool = False,
        loss_weights: Optional[List[float]] = None,
        freeze_conv: bool = False,
        initial_bias: Optional[float] = None,
        dropout: float = 0.25,
        num_conv_layers: int = 16,
        num_nodes: Optional[int] = None,
        graph_pooling: str = "mean",
        edge_dim: Optional[int] = None,
        pe_dim: int = 0,
        global_attn_engine: str = "",
        global_attn_type: str = "",
        global_attn_heads: int = 0,
        use_graph_attr_conditioning: bool = False,
        graph_attr_conditioning_mode: str = "concat_node",
        graph_attr_dim: int = 0,
        var_output: bool = False,
        equivariant_attn_lmax: int = 1,
        equivariant_attn_num_radial: int = 16,
        equivariant_attn_feedforward_multiplier: int = 2,
        equivariant_attn_allow_scalar_only: bool = False,
        equivariant_attn_require_tensor_coupling: bool = True,
        equivariant_attn_chunk_size: Optional[int] = 512,
        equivariant_attn_coupling_mode: str = "parallel",
        **kwargs,
    ):
        super().__init__()
        self.equivariant_attn_lmax = equivariant_attn_lmax
        self.equivariant_attn_num_radial = equivariant_attn_num_radial
        self.equivariant_attn_feedforward_multiplier = \
            equivariant_attn_feedforward_multiplier
        self.equivariant_attn_allow_scalar_only = \
            equivariant_attn_allow_scalar_only
        self.equivariant_attn_require_tensor_coupling = \
            equivariant_attn_require_tensor_coupling
        self.equivariant_attn_chunk_size = equivariant_attn_chunk_size
        self.equivariant_attn_coupling_mode = equivariant_attn_coupling_mode
        self.input_dim = input_dim
        self.hidden_dim = hidden_dim
        self.head_dims = list(output_dim)
        self.head_type = list(output_type)
        self.num_heads = len(self.head_dims)
        self.config_heads = config_heads
        self.equivariance = equivariance
        self.num_conv_layers = num_conv_layers
        self.num_nodes = num_nodes
        self.dropout = dropout
        self.edge_dim = edge_dim
        self.pe_dim = pe_dim
        self.global_attn_engine = global_attn_engine or ""
        self.global_attn_type = global_attn_type or ""
        self.global_attn_heads = global_attn_heads
        self.use_global_attn = bool(self.global_attn_engine) and \
            self.global_attn_engine.lower() not in ("", "none", "false")
        self.conv_checkpointing = False
        # hipGraph-capture safety (train/captured.py): only stacks
        # whose forward has NO data-dependent internal shapes may be
        # replayed across batches (class attr _hipgraph_capture_safe,
        # opt-in per stack).  Dense-batch global attention
        # (to_dense_batch maxN varies per batch) force-disables it.
        self.supports_hipgraph_capture = (
            self._hipgraph_capture_safe and not (
                bool(global_attn_engine)
                and str(global_attn_engine).lower()
                not in ("", "none", "false")))
        self.graph_pooling = graph_pooling
        self.use_graph_attr_conditioning = use_graph_attr_conditioning
        self.graph_attr_conditioning_mode = graph_attr_conditioning_mode.lower()
        self.graph_attr_dim = graph_attr_dim
        if self.graph_attr_conditioning_mode not in (
                "film", "concat_node", "fuse_pool"):
            raise ValueError(
                "graph_attr_conditioning_mode must be film | concat_node"
                " | fuse_pool")

        self.activation_function = activation_function_selection(
            activation_function_type)
        self.loss_function_type = loss_function_type
        self.loss_function = loss_function_selection(loss_function_type)
        self.var_output = 1 if (
            var_output or loss_function_type == "GaussianNLLLoss") else 0
        if loss_weights is None:
            loss_weights = [1.0] * self.num_heads
        self.loss_weights = list(loss_weights)
        assert len(self.loss_weights) == self.num_heads, (
            f"Inconsistent number of loss weights ({len(self.loss_weights)})"
            f" and tasks ({self.num_heads})")

        # subclasses may set these before calling super().__init__ via
        # their own attribute stash; defaults here:
        if not hasattr(self, "is_edge_model"):
            self.is_edge_model = False
        self.use_edge_attr = self.is_edge_model and (edge_dim is not None
                                                     and edge_dim > 0)
        # embed_dim: the dim entering the first conv
        self.embed_dim = input_dim
        self.edge_embed_dim = edge_dim if self.use_edge_attr else None

        # pooling
        pool = graph_pooling.lower()
        if pool == "mean":
            self.pool_fn = global_mean_pool
        elif pool in ("add", "sum"):
            self.pool_fn = global_add_pool
        elif pool == "max":
            self.pool_fn = global_max_pool
        else:
            raise ValueError(f"Unknown graph pooling {graph_pooling}")

        self.graph_convs = ModuleList()
        self.feature_layers = ModuleList()
        self.heads_NN = ModuleList()

        self.is_equivariant_attn = (self.global_attn_engine or "") == \
            "EquivariantTransformer"
        if self.use_global_attn and not self.is_equivariant_attn:
            # GPS positional/feature encoders (reference Base.py:215-260)
            self.pos_emb = Linear(max(self.pe_dim, 1), self.hidden_dim)
            if self.input_dim:
                self.node_emb = Linear(self.input_dim, self.hidden_dim)
                self.node_lin = Linear(2 * self.hidden_dim, self.hidden_dim)
            if self.is_edge_model:
                self.rel_pos_emb = Linear(max(self.pe_dim, 1),
                                          self.hidden_dim)
                if self.use_edge_attr:
                    self.edge_emb = Linear(edge_dim, self.hidden_dim)
                    self.edge_lin = Linear(2 * self.hidden_dim,
                                           self.hidden_dim)
            self.embed_dim = self.hidden_dim
            if self.use_edge_attr or self.is_edge_model:
                self.edge_embed_dim = self.hidden_dim

        self._init_conditioning()
        self._init_conv()
        self._multihead()

        if initial_bias is not None:
            self._set_initial_bias(initial_bias)
        if freeze_conv:
            self._freeze_conv()

    # ------------------------------------------------------------------
    # construction
    # ------------------------------------------------------------------
    def get_conv(self, input_dim, output_dim, edge_dim=None):
        raise NotImplementedError

    def _apply_global_attn(self, conv):
        """Wrap a local conv in a GPS or EquivariantTransformer layer
        when global attention is on (reference Base.py:264)."""
        if not self.use_global_attn:
            return conv
        if self.is_equivariant_attn:
            from ..globalatt.equivariant import (
                EquivariantLocalGlobalConv,
                EquivariantTransformerLayer,
                create_local_feature_adapter,
            )
            mpnn = str(self).replace("Stack", "").replace("SCF", "SchNet")
            lmax = self.equivariant_attn_lmax
            adapter = create_local_feature_adapter(
                mpnn, self.hidden_dim, self.hidden_dim, lmax,
                allow_scalar_only=self.equivariant_attn_allow_scalar_only)
            layer = EquivariantTransformerLayer(
                self.hidden_dim, lmax,
                num_heads=self.global_attn_heads or 4,
                num_radial=self.equivariant_attn_num_radial,
                feedforward_multiplier=
                self.equivariant_attn_feedforward_multiplier,
                chunk_size=self.equivariant_attn_chunk_size)
            return EquivariantLocalGlobalConv(
                conv, adapter, layer,
                mode=self.equivariant_attn_coupling_mode)
        from ..globalatt.gps import HydraGPSConv
        return HydraGPSConv(
            self.hidden_dim, conv, heads=self.global_attn_heads or 1,
            dropout=self.dropout,
            attn_type=(self.global_attn_type or "multihead"))

    def _init_conv(self):
        self.graph_convs.append(self._apply_global_attn(
            self.get_conv(self.embed_dim, self.hidden_dim,
                          edge_dim=self.edge_embed_dim)))
        self.feature_layers.append(BatchNormNode(self.hidden_dim))
        for _ in range(self.num_conv_layers - 1):
            self.graph_convs.append(self._apply_global_attn(
                self.get_conv(self.hidden_dim, self.hidden_dim,
                              edge_dim=self.edge_embed_dim)))
            self.feature_layers.append(BatchNormNode(self.hidden_dim))

    def _init_conditioning(self):
        if not self.use_graph_attr_conditioning:
            return
        g = max(self.graph_attr_dim, 1)
        if self.graph_attr_conditioning_mode == "film":
            self.film = Linear(g, 2 * self.hidden_dim)
        elif self.graph_attr_conditioning_mode == "concat_node":
            self.node_attr_projector = Linear(self.hidden_dim + g,
                                              self.hidden_dim)
        else:  # fuse_pool
            self.graph_pool_projector = Linear(self.hidden_dim + g,
                                               self.hidden_dim)

    def _init_node_conv(self):
        """Per-branch conv decoders for node heads of type 'conv'."""
        node_branches = self.config_heads.get("node", [])
        use_conv = any(b["architecture"]["type"] == "conv"
                       for b in node_branches)
        if not use_conv:
            return
        self.convs_node_hidden = ModuleDict({})
        self.batch_norms_node_hidden = ModuleDict({})
        self.convs_node_output = ModuleDict({})
        self.batch_norms_node_output = ModuleDict({})
        node_head_dims = [d for d, t in zip(self.head_dims, self.head_type)
                          if t == "node"]
        for branchdict in node_branches:
            if branchdict["architecture"]["type"] != "conv":
                continue
            btype = branchdict["type"]
            hidden_dims = branchdict["architecture"]["dim_headlayers"]
            convs = ModuleList()
            bns = ModuleList()
            convs.append(self.get_conv(self.hidden_dim, hidden_dims[0]))
            bns.append(BatchNormNode(hidden_dims[0]))
            for i in range(len(hidden_dims) - 1):
                convs.append(self.get_conv(hidden_dims[i], hidden_dims[i + 1]))
                bns.append(BatchNormNode(hidden_dims[i + 1]))
            self.convs_node_hidden[btype] = convs
            self.batch_norms_node_hidden[btype] = bns
            out_convs = ModuleList()
            out_bns = ModuleList()
            for hd in node_head_dims:
                out_convs.append(
                    self.get_conv(hidden_dims[-1], hd * (1 + self.var_output)))
                out_bns.append(BatchNormNode(hd * (1 + self.var_output)))
            self.convs_node_output[btype] = out_convs
            self.batch_norms_node_output[btype] = out_bns

    def _multihead(self):
        self.graph_shared = ModuleDict({})
        self.num_branches = 1
        if "graph" in self.config_heads:
            self.num_branches = len(self.config_heads["graph"])
            if self.num_branches > 1:
                # multi-branch forwards partition by data.dataset_name
                # with boolean masks (data-dependent shapes) — never
                # replay them from a captured hipGraph
                self.supports_hipgraph_capture = False
            for branchdict in self.config_heads["graph"]:
                arch = branchdict["architecture"]
                dim_shared = arch["dim_sharedlayers"]
                layers = [Linear(self.hidden_dim, dim_shared),
                          self.activation_function]
                for _ in range(arch["num_sharedlayers"] - 1):
                    layers += [Linear(dim_shared, dim_shared),
                               self.activation_function]
                self.graph_shared[branchdict["type"]] = Sequential(*layers)

        if "node" in self.config_heads:
            self._init_node_conv()

        inode_feature = 0
        for ihead in range(self.num_heads):
            head_NN = ModuleDict({})
            if self.head_type[ihead] == "graph":
                for branchdict in self.config_heads["graph"]:
                    arch = branchdict["architecture"]
                    dims = arch["dim_headlayers"]
                    layers = [Linear(arch["dim_sharedlayers"], dims[0]),
                              self.activation_function]
                    for i in range(arch["num_headlayers"] - 1):
                        layers += [Linear(dims[i], dims[i + 1]),
                                   self.activation_function]
                    layers.append(
                        Linear(dims[-1],
                               self.head_dims[ihead] * (1 + self.var_output)))
                    head_NN[branchdict["type"]] = Sequential(*layers)
            elif self.head_type[ihead] == "node":
                for branchdict in self.config_heads["node"]:
                    btype = branchdict["type"]
                    arch = branchdict["architecture"]
                    node_NN_type = arch["type"]
                    if node_NN_type in ("mlp", "mlp_per_node"):
                        num_mlp = 1 if node_NN_type == "mlp" else self.num_nodes
                        head_NN[btype] = MLPNode(
                            self.hidden_dim,
                            self.head_dims[ihead] * (1 + self.var_output),
                            num_mlp, arch["dim_headlayers"], node_NN_type,
                            self.activation_function,
                            num_nodes=self.num_nodes
                            if node_NN_type == "mlp_per_node" else None)
                    elif node_NN_type == "conv":
                        mods = ModuleList()
                        for conv, bn in zip(self.convs_node_hidden[btype],
                                            self.batch_norms_node_hidden[btype]):
                            mods.append(conv)
                            mods.append(bn)
                        mods.append(self.convs_node_output[btype][inode_feature])
                        mods.append(
                            self.batch_norms_node_output[btype][inode_feature])
                        head_NN[btype] = mods
                    else:
                        raise ValueError(
                            f"Unknown node head type {node_NN_type}")
                if self.head_type[ihead] == "node" and any(
                        b["architecture"]["type"] == "conv"
                        for b in self.config_heads["node"]):
                    inode_feature += 1
            else:
                raise ValueError(f"Unknown head type {self.head_type[ihead]}")
            self.heads_NN.append(head_NN)

    def _set_initial_bias(self, bias):
        for head in self.heads_NN:
            for mod in head.values():
                last = None
                for m in mod.modules():
                    if isinstance(m, Linear):
                        last = m
                if last is not None and last.bias is not None:
                    nn.init.constant_(last.bias, bias)

    def _freeze_conv(self):
        for module in [self.graph_convs, self.feature_layers]:
            for p in module.parameters():
                p.requires_grad = False

    def enable_conv_checkpointing(self):
        self.conv_checkpointing = True
        # torch checkpointing's recompute-in-backward + RNG stashing
        # is not replayable from a captured hipGraph
        self.supports_hipgraph_capture = False

    # ------------------------------------------------------------------
    # forward
    # ------------------------------------------------------------------
    def _embedding(self, data):
        """Returns (inv_node_feat, equiv_node_feat, conv_args).
        Subclasses with geometry override this."""
        conv_args = {"edge_index": data.edge_index}
        if self.use_edge_attr:
            assert data.get("edge_attr") is not None, \
                "Data must have edge attributes if use_edge_attr is set"
            conv_args["edge_attr"] = data.edge_attr
        x = data.x
        if x is not None and x.dtype not in (torch.float32, torch.float64,
                                             torch.bfloat16, torch.float16):
            x = x.float()
        pos = data.get("pos")
        if self.use_global_attn and self.is_equivariant_attn:
            conv_args["batch"] = data.get("batch")
            return x, pos, conv_args
        if self.use_global_attn:
            h, conv_args = self._gps_encode(data, x, conv_args)
            return h, pos, conv_args
        return x, pos, conv_args

    def _gps_encode(self, data, x, conv_args):
        """GPS positional/feature encoders (reference Base.py:523-557);
        reusable by stacks that override _embedding."""
        conv_args["batch"] = data.get("batch")
        dt = self.pos_emb.weight.dtype
        pe = data.get("pe")
        if pe is None:
            raise ValueError(
                "GPS global attention requires positional encodings "
                "(data.pe); add them in preprocessing "
                "(hydragnn_amd.preprocess.add_laplacian_pe)")
        h = self.pos_emb(pe.to(dt))
        if self.input_dim:
            h = self.node_lin(
                torch.cat([self.node_emb(x.to(dt)), h], dim=1))
        if self.is_edge_model:
            rel_pe = data.get("rel_pe")
            if rel_pe is None:
                src, dst = data.edge_index[0], data.edge_index[1]
                rel_pe = (pe[dst] - pe[src]).abs()
            e = self.rel_pos_emb(rel_pe.to(dt))
            if self.use_edge_attr:
                e = self.edge_lin(torch.cat(
                    [self.edge_emb(conv_args["edge_attr"]), e], dim=1))
            conv_args["edge_attr"] = e
        return h, conv_args

    def _apply_graph_conditioning(self, x, batch, data):
        if not self.use_graph_attr_conditioning:
            return x
        ga = data.get("graph_attr")
        if ga is None:
            return x
        if ga.dim() == 1:
            ga = ga.view(-1, 1)
        ga = ga.to(x.dtype)
        if batch is None:
            ga_nodes = ga.expand(x.shape[0], -1)
        else:
            ga_nodes = ga[batch]
        if self.graph_attr_conditioning_mode == "film":
            gamma_beta = self.film(ga_nodes)
            gamma, beta = gamma_beta.chunk(2, dim=-1)
            return x * (1.0 + gamma) + beta
        if self.graph_attr_conditioning_mode == "concat_node":
            return self.node_attr_projector(torch.cat([x, ga_nodes], dim=-1))
        return x  # fuse_pool handled at pooling time

    def _apply_graph_pool_conditioning(self, x_graph, data):
        if (not self.use_graph_attr_conditioning
                or self.graph_attr_conditioning_mode != "fuse_pool"):
            return x_graph
        ga = data.get("graph_attr")
        if ga is None:
            return x_graph
        if ga.dim() == 1:
            ga = ga.view(-1, 1)
        return self.graph_pool_projector(
            torch.cat([x_graph, ga.to(x_graph.dtype)], dim=-1))

    def _conv_block(self, conv, feat_layer, inv, equiv, batch, data,
                    conv_args):
        if self.conv_checkpointing:
            inv, equiv = checkpoint(conv, use_reentrant=False,
                                    inv_node_feat=inv, equiv_node_feat=equiv,
                                    **conv_args)
        else:
            inv, equiv = conv(inv_node_feat=inv, equiv_node_feat=equiv,
                              **conv_args)
        inv = self._apply_graph_conditioning(inv, batch, data)
        inv = self.activation_function(feat_layer(inv))
        return inv, equiv

    def encoder_forward(self, data):
        """Embedding + conv stack + pooling (the shared encoder half
        used by MultiTaskModelMP)."""
        # propagate the collation-time dst-sortedness flag to every
        # conv so edge aggregations take the deterministic CSR kernel
        # instead of atomics (host-side attr; capture-safe)
        es = bool(data.get("edges_sorted_", False))
        if getattr(self, "_edges_sorted_flag", None) != es:
            for m in self.modules():
                m._edges_sorted = es
            self._edges_sorted_flag = es
        inv_node_feat, equiv_node_feat, conv_args = self._embedding(data)
        batch = data.get("batch")

        for conv, feat_layer in zip(self.graph_convs, self.feature_layers):
            inv_node_feat, equiv_node_feat = self._conv_block(
                conv, feat_layer, inv_node_feat, equiv_node_feat, batch,
                data, conv_args)

        x = inv_node_feat
        if batch is None:
            batch = torch.zeros(x.shape[0], dtype=torch.long, device=x.device)
            data["batch"] = batch
        n_graphs = data.get("num_graphs_")
        if n_graphs is None:
            n_graphs = int(batch.max()) + 1 if batch.numel() > 0 else 1
        x_graph = self.pool_fn(x, batch, int(n_graphs))
        x_graph = self._apply_graph_pool_conditioning(x_graph, data)
        return x, x_graph, equiv_node_feat, conv_args

    def forward(self, data):
        x, x_graph, equiv_node_feat, conv_args = self.encoder_forward(data)
        return self._decode(x, x_graph, equiv_node_feat, data, conv_args)

    def _decode(self, x, x_graph, equiv_node_feat, data, conv_args):
        batch = data["batch"]
        outputs = []
        outputs_var = []
        dataset_name = data.get("dataset_name")
        if dataset_name is None:
            dataset_name = torch.zeros(
                x_graph.shape[0], 1, dtype=torch.long, device=x.device)
            data["dataset_name"] = dataset_name
        if dataset_name.dim() == 1:
            dataset_name = dataset_name.view(-1, 1)
        if self.num_branches > 1:
            # device->host sync is acceptable only on the multi-branch
            # path; the single-branch hot path stays hipGraph-capturable
            datasetIDs = dataset_name.unique()
            _, node_counts = torch.unique_consecutive(
                batch, return_counts=True)
        else:
            datasetIDs = None
            node_counts = None

        for head_dim, headloc, type_head in zip(self.head_dims, self.heads_NN,
                                                self.head_type):
            if type_head == "graph":
                if self.num_branches == 1:
                    xh = self.graph_shared["branch-0"](x_graph)
                    out = headloc["branch-0"](xh)
                    head = out[:, :head_dim]
                    headvar = out[:, head_dim:] ** 2
                else:
                    head = x_graph.new_zeros(x_graph.shape[0], head_dim)
                    headvar = x_graph.new_zeros(
                        x_graph.shape[0], head_dim * self.var_output)
                    for ID in datasetIDs:
                        mask = (dataset_name == ID)[:, 0]
                        btype = f"branch-{int(ID)}"
                        xh = self.graph_shared[btype](x_graph[mask])
                        out = headloc[btype](xh)
                        head = head.index_put(
                            (mask.nonzero(as_tuple=True)[0],),
                            out[:, :head_dim])
                        if self.var_output:
                            headvar = headvar.index_put(
                                (mask.nonzero(as_tuple=True)[0],),
                                out[:, head_dim:] ** 2)
                outputs.append(head)
                outputs_var.append(headvar if self.var_output else None)
            else:
                node_NN_type = self.config_heads["node"][0]["architecture"]["type"]
                if self.num_branches == 1:
                    x_node = self._node_head_forward(
                        headloc["branch-0"], node_NN_type, x, equiv_node_feat,
                        batch, conv_args)
                    head = x_node[:, :head_dim]
                    headvar = x_node[:, head_dim:] ** 2
                else:
                    head = x.new_zeros(x.shape[0], head_dim)
                    headvar = x.new_zeros(
                        x.shape[0], head_dim * self.var_output)
                    for ID in datasetIDs:
                        mask = (dataset_name == ID)[:, 0]
                        mask_nodes = torch.repeat_interleave(mask, node_counts)
                        btype = f"branch-{int(ID)}"
                        x_node = self._node_head_forward(
                            headloc[btype], node_NN_type, x[mask_nodes],
                            None if equiv_node_feat is None
                            else equiv_node_feat[mask_nodes],
                            batch[mask_nodes], conv_args)
                        nidx = mask_nodes.nonzero(as_tuple=True)[0]
                        head = head.index_put((nidx,), x_node[:, :head_dim])
                        if self.var_output:
                            headvar = headvar.index_put(
                                (nidx,), x_node[:, head_dim:] ** 2)
                outputs.append(head)
                outputs_var.append(headvar if self.var_output else None)
        if self.var_output:
            return outputs, outputs_var
        return outputs

    def _node_head_forward(self, head_mod, node_NN_type, x, equiv, batch,
                           conv_args):
        if node_NN_type == "conv":
            inv = x
            eq = equiv
            for conv, bn in zip(head_mod[0::2], head_mod[1::2]):
                inv, eq = conv(inv_node_feat=inv, equiv_node_feat=eq,
                               **conv_args)
                inv = self.activation_function(bn(inv))
            return inv
        return head_mod(x=x, batch=batch)

    # ------------------------------------------------------------------
    # loss
    # ------------------------------------------------------------------
    def loss(self, pred, value, head_index):
        var = None
        if self.var_output:
            pred, var = pred
        return self.loss_hpweighted(pred, value, head_index, var=var)

    def loss_hpweighted(self, pred, value, head_index, var=None):
        tot_loss = 0
        tasks_loss = []
        for ihead in range(self.num_heads):
            head_pre = pred[ihead]
            head_val = value[head_index[ihead]]
            if head_pre.shape != head_val.shape:
                head_val = head_val.reshape(head_pre.shape)
            if var is None:
                l = self.loss_function(head_pre, head_val)
            else:
                l = self.loss_function(head_pre, head_val, var[ihead])
            tot_loss = tot_loss + l * self.loss_weights[ihead]
            tasks_loss.append(l.detach())
        return tot_loss, tasks_loss

    def __str__(self):
        return "Base"


class MLPNode(Module):
    """Node-level MLP decoder; 'mlp' shares one MLP over all nodes,
    'mlp_per_node' has one per node slot (fixed-size graphs)."""

    def __init__(self, input_dim, output_dim, num_mlp, hidden_dim_node,
                 node_type, activation_function, num_nodes=None):
        super().__init__()
        self.input_dim = input_dim
        self.output_dim = output_dim
        self.node_type = node_type
        self.num_mlp = num_mlp
        self.num_nodes = num_nodes
        self.activation_function = activation_function
        self.mlp = ModuleList()
        for _ in range(num_mlp):
            layers = [Linear(input_dim, hidden_dim_node[0]),
                      activation_function]
            for i in range(len(hidden_dim_node) - 1):
                layers += [Linear(hidden_dim_node[i], hidden_dim_node[i + 1]),
                           activation_function]
            layers.append(Linear(hidden_dim_node[-1], output_dim))
            self.mlp.append(Sequential(*layers))

    def forward(self, x: torch.Tensor, batch: torch.Tensor):
        if self.node_type == "mlp":
            return self.mlp[0](x)
        assert self.num_nodes is not None
        outs = torch.zeros(x.shape[0], self.output_dim, dtype=x.dtype,
                           device=x.device)
        for inode in range(self.num_nodes):
            idx = torch.arange(inode, batch.shape[0], self.num_nodes,
                               device=x.device)
            outs[idx] = self.mlp[inode](x[idx])
        return outs

    def __str__(self):
        return "MLPNode"
