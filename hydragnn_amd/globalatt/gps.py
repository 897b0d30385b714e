"""GraphGPS global-attention layer.

Behavioral parity with /root/reference/hydragnn/globalAtt/gps.py:30-227:
local MPNN branch + dense-batched global attention (torch
MultiheadAttention or Performer/FAVOR+ linear attention) + MLP, with
residuals and three norms; Performer projection redraw is driven by the
training loop, not forward (safe under activation checkpointing).

On MI355X the multihead path runs the segment-varlen HIP kernel
(ops/csrc/varlen_attn.hip): one (graph, head) workgroup with K/V in
LDS and an online softmax over exactly the graph's nodes — no dense
padding.  Batches outside its envelope (head_dim > 32 or a graph with
> 256 nodes) and the Performer path fall back to dense-batch SDPA.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
from torch import nn

from ..data import to_dense_batch


class PerformerAttention(nn.Module):
    """FAVOR+ linear attention (Choromanski et al. 2021), batch-dense
    with key padding mask; redraw() resamples the random projection."""

    def __init__(self, channels: int, heads: int, num_features: int = 64):
        super().__init__()
        assert channels % heads == 0
        self.heads = heads
        self.head_dim = channels // heads
        self.num_features = num_features
        self.qkv = nn.Linear(channels, 3 * channels)
        self.out = nn.Linear(channels, channels)
        self.register_buffer(
            "projection", self._draw(self.num_features, self.head_dim))

    @staticmethod
    def _draw(m: int, d: int) -> torch.Tensor:
        # orthogonal random features
        blocks = []
        for _ in range(math.ceil(m / d)):
            q, _ = torch.linalg.qr(torch.randn(d, d))
            blocks.append(q.t())
        proj = torch.cat(blocks, dim=0)[:m]
        norms = torch.randn(m, d).norm(dim=1, keepdim=True)
        return proj * norms

    def redraw_projection_matrix(self):
        self.projection.copy_(
            self._draw(self.num_features, self.head_dim).to(
                self.projection.device))

    def _phi(self, x: torch.Tensor) -> torch.Tensor:
        # x [B, H, N, d] -> positive random features [B, H, N, m]
        proj = self.projection.to(x.dtype)  # [m, d]
        xp = x @ proj.t() / (self.head_dim ** 0.25)
        x_norm = (x ** 2).sum(-1, keepdim=True) / (
            2 * math.sqrt(self.head_dim))
        return torch.exp(xp - x_norm - xp.amax(dim=-1, keepdim=True)
                         ) + 1e-6

    def forward(self, x: torch.Tensor, mask: Optional[torch.Tensor] = None):
        B, N, C = x.shape
        H, d = self.heads, self.head_dim
        q, k, v = self.qkv(x).chunk(3, dim=-1)
        q = q.view(B, N, H, d).transpose(1, 2)
        k = k.view(B, N, H, d).transpose(1, 2)
        v = v.view(B, N, H, d).transpose(1, 2)
        qp = self._phi(q)
        kp = self._phi(k)
        if mask is not None:
            kp = kp * mask.view(B, 1, N, 1)
            v = v * mask.view(B, 1, N, 1)
        kv = torch.einsum("bhnm,bhnd->bhmd", kp, v)
        z = 1.0 / (torch.einsum("bhnm,bhm->bhn", qp,
                                kp.sum(dim=2)) + 1e-6)
        out = torch.einsum("bhnm,bhmd,bhn->bhnd", qp, kv, z)
        out = out.transpose(1, 2).reshape(B, N, C)
        return self.out(out)


class HydraGPSConv(nn.Module):
    """Local MPNN + global attention + MLP with residuals/norms, on the
    two-stream (inv, equiv) interface."""

    def __init__(self, channels: int, conv: Optional[nn.Module],
                 heads: int = 1, dropout: float = 0.0,
                 attn_type: str = "multihead"):
        super().__init__()
        self.channels = channels
        self.conv = conv
        self.heads = heads
        self.dropout = dropout
        self.attn_type = attn_type
        if attn_type == "multihead":
            self.attn = nn.MultiheadAttention(channels, heads,
                                              batch_first=True)
        elif attn_type == "performer":
            self.attn = PerformerAttention(channels, heads)
        else:
            raise ValueError(f"attn_type {attn_type} not supported")
        self.mlp = nn.Sequential(
            nn.Linear(channels, channels * 2), nn.ReLU(),
            nn.Dropout(dropout), nn.Linear(channels * 2, channels),
            nn.Dropout(dropout))
        self.norm1 = nn.BatchNorm1d(channels)
        self.norm2 = nn.BatchNorm1d(channels)
        self.norm3 = nn.BatchNorm1d(channels)
        self._performer_steps_since_redraw = 0

    def redraw_projection(self, redraw_interval: Optional[int]) -> bool:
        if (self.attn_type != "performer" or redraw_interval is None
                or not self.training):
            return False
        self._performer_steps_since_redraw += 1
        if self._performer_steps_since_redraw >= redraw_interval:
            self.attn.redraw_projection_matrix()
            self._performer_steps_since_redraw = 0
            return True
        return False

    def _try_varlen_attention(self, x, batch):
        """Run nn.MultiheadAttention's math through the segment-varlen
        HIP kernel (no dense padding).  Returns None when the batch
        does not fit the kernel envelope (falls back to dense SDPA)."""
        from ..ops.scatter import _rowptr_from_sorted
        from ..ops.varlen_attn import varlen_attention, varlen_eligible
        attn = self.attn
        head_dim = self.channels // self.heads
        if not varlen_eligible(head_dim, 0, x.device):
            return None
        if attn.in_proj_weight is None or not attn.batch_first:
            return None
        num_graphs = int(batch[-1]) + 1 if batch.numel() else 0
        # (r2: the K/V-tiled kernel has no segment-length cap, so no
        # device sync to check max_seg)
        ptr = _rowptr_from_sorted(batch, num_graphs)
        qkv = nn.functional.linear(x, attn.in_proj_weight,
                                   attn.in_proj_bias)
        q, k, v = qkv.view(x.shape[0], 3, self.heads,
                           head_dim).unbind(dim=1)
        out = varlen_attention(q, k, v, ptr, batch)
        return attn.out_proj(out.reshape(x.shape[0], self.channels))

    def forward(self, inv_node_feat, equiv_node_feat, batch=None, **kwargs):
        hs = []
        equiv_out = equiv_node_feat
        if self.conv is not None:
            h_local, equiv_out = self.conv(
                inv_node_feat=inv_node_feat,
                equiv_node_feat=equiv_node_feat, **kwargs)
            h_local = nn.functional.dropout(h_local, self.dropout,
                                            self.training)
            h_local = h_local + inv_node_feat
            h_local = self.norm1(h_local)
            hs.append(h_local)

        # global attention: segment-varlen HIP kernel when the batch
        # fits its envelope (molecular graphs), dense batch otherwise
        h_attn = None
        if self.attn_type == "multihead" and batch is not None:
            h_attn = self._try_varlen_attention(inv_node_feat, batch)
        if h_attn is None:
            x_dense, mask = to_dense_batch(inv_node_feat, batch)
            if self.attn_type == "multihead":
                h_attn, _ = self.attn(x_dense, x_dense, x_dense,
                                      key_padding_mask=~mask,
                                      need_weights=False)
            else:
                h_attn = self.attn(x_dense, mask=mask)
            h_attn = h_attn[mask]
        h_attn = nn.functional.dropout(h_attn, self.dropout, self.training)
        h_attn = h_attn + inv_node_feat
        h_attn = self.norm2(h_attn)
        hs.append(h_attn)

        out = sum(hs)
        out = out + self.mlp(out)
        out = self.norm3(out)
        return out, equiv_out


def redraw_performer_projections(model, redraw_interval: Optional[int]):
    """Training-loop hook: walk the module tree and redraw due
    Performer projections (reference gps.py:214)."""
    m = model.module if hasattr(model, "module") else model
    redrawn = False
    for mod in m.modules():
        if isinstance(mod, HydraGPSConv):
            redrawn |= mod.redraw_projection(redraw_interval)
    return redrawn
