"""DimeNet++ stack: directional message passing.

Re-implementation of the DimeNet++ operator (Gasteiger et al. 2020)
with the reference's capability surface (/root/reference/hydragnn/
models/DIMEStack.py:34-355): triplet enumeration via bincount/argsort
(no torch_sparse), angle computation atan2(|ji x ki|, ji.ki), Bessel
RBF + spherical Bessel x Legendre SBF (zeros computed at init with
scipy), InteractionPP / OutputPP blocks.
"""

from __future__ import annotations

import math
from typing import Optional

import numpy as np
import torch
from torch import nn

from ..ops import gather, get_edge_vectors_and_lengths, scatter
from .base import Base
from .pna_plus import BesselBasisLayer


def triplets(edge_index: torch.Tensor, num_nodes: int):
    """For each edge j->i enumerate edges k->j (k != i).
    Returns (idx_kj, idx_ji): edge ids of the (k->j, j->i) pairs.
    Pure index arithmetic (bincount / argsort), mirroring the
    reference's AMD-safe path (DIMEStack.py:260-306) but vectorized."""
    src, dst = edge_index[0], edge_index[1]
    E = src.shape[0]
    dev = edge_index.device
    # group edges by destination
    perm = torch.argsort(dst, stable=True)
    sorted_dst = dst[perm]
    counts = torch.bincount(sorted_dst, minlength=num_nodes)
    rowptr = torch.zeros(num_nodes + 1, dtype=torch.long, device=dev)
    rowptr[1:] = counts.cumsum(0)
    # edge e has source j = src[e]; predecessors = edges with dst == j
    n_trip = counts[src]  # per-edge number of incoming edges at j
    idx_ji = torch.repeat_interleave(torch.arange(E, device=dev), n_trip)
    # position within each group
    offs = rowptr[src[idx_ji]]
    cum = torch.cumsum(n_trip, 0) - n_trip
    pos_in_group = torch.arange(idx_ji.shape[0], device=dev) - cum[idx_ji]
    idx_kj = perm[offs + pos_in_group]
    # drop k == i (backtracking triplets)
    mask = src[idx_kj] != dst[idx_ji]
    return idx_kj[mask], idx_ji[mask]


def _spherical_bessel_zeros(num_spherical: int, num_radial: int) -> np.ndarray:
    from scipy import special, optimize
    zeros = np.zeros((num_spherical, num_radial))
    # l = 0: zeros of sin(x)/x are n*pi
    zeros[0] = np.arange(1, num_radial + 1) * math.pi
    pts = np.arange(1, num_radial + num_spherical + 1) * math.pi
    prev = pts  # zeros of j_0 (enough of them)
    for l in range(1, num_spherical):
        f = lambda x: special.spherical_jn(l, x)
        cur = []
        # zeros of j_l interlace those of j_{l-1}
        for a, b in zip(prev[:-1], prev[1:]):
            cur.append(optimize.brentq(f, a + 1e-9, b - 1e-9))
        prev = np.array(cur)
        zeros[l] = prev[:num_radial]
    return zeros


def _spherical_jn_torch(l: int, x: torch.Tensor) -> torch.Tensor:
    """Upward recursion, safe near 0."""
    xs = x.clamp(min=1e-7)
    j0 = torch.sin(xs) / xs
    if l == 0:
        return j0
    j1 = torch.sin(xs) / xs ** 2 - torch.cos(xs) / xs
    if l == 1:
        return j1
    jm, jc = j0, j1
    for ll in range(1, l):
        jn = (2 * ll + 1) / xs * jc - jm
        jm, jc = jc, jn
    return jc


def _legendre_torch(l: int, x: torch.Tensor) -> torch.Tensor:
    if l == 0:
        return torch.ones_like(x)
    if l == 1:
        return x
    pm, pc = torch.ones_like(x), x
    for ll in range(1, l):
        pn = ((2 * ll + 1) * x * pc - ll * pm) / (ll + 1)
        pm, pc = pc, pn
    return pc


class SphericalBasisLayer(nn.Module):
    def __init__(self, num_spherical: int, num_radial: int, cutoff: float,
                 envelope_exponent: int = 5):
        super().__init__()
        self.num_spherical = num_spherical
        self.num_radial = num_radial
        self.cutoff = cutoff
        self.envelope_exponent = envelope_exponent
        zeros = _spherical_bessel_zeros(num_spherical, num_radial)
        self.register_buffer("zeros", torch.from_numpy(zeros).float())
        # normalization sqrt(2 / (c^3 j_{l+1}(z_ln)^2))
        from scipy import special
        norms = np.zeros_like(zeros)
        for l in range(num_spherical):
            norms[l] = np.sqrt(
                2.0 / (cutoff ** 3
                       * special.spherical_jn(l + 1, zeros[l]) ** 2))
        self.register_buffer("norms", torch.from_numpy(norms).float())

    def envelope(self, x):
        p = self.envelope_exponent + 1
        a = -(p + 1) * (p + 2) / 2
        b = p * (p + 2)
        c = -p * (p + 1) / 2
        xc = x.clamp(min=1e-9)
        return (1.0 / xc + a * x.pow(p - 1) + b * x.pow(p)
                + c * x.pow(p + 1)) * (x < 1.0).to(x.dtype)

    def forward(self, dist, angle, idx_kj):
        """dist: [E] edge lengths; angle: [T]; idx_kj: [T] edge id of
        the kj edge -> output [T, num_spherical*num_radial]."""
        d = dist / self.cutoff  # [E]
        d_kj = d[idx_kj]  # [T]
        env = self.envelope(d_kj).unsqueeze(-1)
        cos_a = torch.cos(angle)
        outs = []
        for l in range(self.num_spherical):
            radial = _spherical_jn_torch(
                l, self.zeros[l].view(1, -1).to(d.dtype)
                * d_kj.unsqueeze(-1))  # [T, num_radial]
            radial = radial * self.norms[l].view(1, -1).to(d.dtype)
            ang = _legendre_torch(l, cos_a) * math.sqrt(
                (2 * l + 1) / (4 * math.pi))
            outs.append(env * radial * ang.unsqueeze(-1))
        return torch.cat(outs, dim=-1)


class HydraEmbeddingBlock(nn.Module):
    def __init__(self, in_dim: int, hidden: int, num_radial: int):
        super().__init__()
        self.lin_x = nn.Linear(in_dim, hidden)
        self.lin_rbf = nn.Linear(num_radial, hidden)
        self.lin = nn.Linear(3 * hidden, hidden)
        self.act = nn.SiLU()

    def forward(self, x, rbf, edge_index):
        src, dst = edge_index[0], edge_index[1]
        h = self.act(self.lin_x(x))
        return self.act(self.lin(torch.cat(
            [gather(h, src), gather(h, dst),
             self.act(self.lin_rbf(rbf))], dim=-1)))


class ResidualLayer(nn.Module):
    def __init__(self, hidden: int):
        super().__init__()
        self.lin1 = nn.Linear(hidden, hidden)
        self.lin2 = nn.Linear(hidden, hidden)
        self.act = nn.SiLU()

    def forward(self, x):
        return x + self.act(self.lin2(self.act(self.lin1(x))))


class InteractionPPBlock(nn.Module):
    def __init__(self, hidden: int, int_emb_size: int, basis_emb_size: int,
                 num_radial: int, num_spherical: int,
                 num_before_skip: int, num_after_skip: int):
        super().__init__()
        self.act = nn.SiLU()
        self.lin_rbf1 = nn.Linear(num_radial, basis_emb_size, bias=False)
        self.lin_rbf2 = nn.Linear(basis_emb_size, hidden, bias=False)
        self.lin_sbf1 = nn.Linear(num_spherical * num_radial,
                                  basis_emb_size, bias=False)
        self.lin_sbf2 = nn.Linear(basis_emb_size, int_emb_size, bias=False)
        self.lin_kj = nn.Linear(hidden, hidden)
        self.lin_ji = nn.Linear(hidden, hidden)
        self.lin_down = nn.Linear(hidden, int_emb_size, bias=False)
        self.lin_up = nn.Linear(int_emb_size, hidden, bias=False)
        self.before_skip = nn.ModuleList(
            [ResidualLayer(hidden) for _ in range(num_before_skip)])
        self.lin = nn.Linear(hidden, hidden)
        self.after_skip = nn.ModuleList(
            [ResidualLayer(hidden) for _ in range(num_after_skip)])

    def forward(self, m, rbf, sbf, idx_kj, idx_ji):
        x_ji = self.act(self.lin_ji(m))
        x_kj = self.act(self.lin_kj(m))
        x_kj = x_kj * self.lin_rbf2(self.lin_rbf1(rbf))
        x_kj = self.act(self.lin_down(x_kj))
        x_kj = gather(x_kj, idx_kj) * self.lin_sbf2(self.lin_sbf1(sbf))
        x_kj = scatter(x_kj, idx_ji, m.shape[0], "sum")
        x_kj = self.act(self.lin_up(x_kj))
        h = x_ji + x_kj
        for layer in self.before_skip:
            h = layer(h)
        h = self.act(self.lin(h)) + m
        for layer in self.after_skip:
            h = layer(h)
        return h


class OutputPPBlock(nn.Module):
    def __init__(self, hidden: int, out_emb: int, out_dim: int,
                 num_radial: int):
        super().__init__()
        self.act = nn.SiLU()
        self.lin_rbf = nn.Linear(num_radial, hidden, bias=False)
        self.lin_up = nn.Linear(hidden, out_emb, bias=False)
        self.lins = nn.ModuleList([nn.Linear(out_emb, out_emb)
                                   for _ in range(2)])
        self.lin_out = nn.Linear(out_emb, out_dim)

    def forward(self, m, rbf, edge_index, num_nodes):
        dst = edge_index[1]
        w = self.lin_rbf(rbf) * m
        x = scatter(w, dst, num_nodes, "sum")
        x = self.lin_up(x)
        for lin in self.lins:
            x = self.act(lin(x))
        return self.lin_out(x)


class _DimeConv(nn.Module):
    """One interaction + output block: edge messages in, node feats out."""

    def __init__(self, hidden, int_emb, basis_emb, out_emb, out_dim,
                 num_radial, num_spherical, nbs, nas):
        super().__init__()
        self.interaction = InteractionPPBlock(
            hidden, int_emb, basis_emb, num_radial, num_spherical, nbs, nas)
        self.output = OutputPPBlock(hidden, out_emb, out_dim, num_radial)

    def forward(self, inv_node_feat, equiv_node_feat, edge_index, rbf, sbf,
                idx_kj, idx_ji, msg_state, **kwargs):
        m = msg_state["m"]
        m = self.interaction(m, rbf, sbf, idx_kj, idx_ji)
        msg_state["m"] = m
        x = self.output(m, rbf, edge_index, inv_node_feat.shape[0])
        return x, equiv_node_feat


class DIMEStack(Base):
    def __init__(self, basis_emb_size=8, envelope_exponent=5,
                 int_emb_size=64, out_emb_size=128, num_after_skip=2,
                 num_before_skip=1, num_radial=6, num_spherical=7,
                 edge_dim: Optional[int] = None, radius: float = 5.0,
                 **kwargs):
        self.basis_emb_size = basis_emb_size or 8
        self.envelope_exponent = envelope_exponent or 5
        self.int_emb_size = int_emb_size or 64
        self.out_emb_size = out_emb_size or 128
        self.num_after_skip = num_after_skip if num_after_skip is not None else 2
        self.num_before_skip = num_before_skip if num_before_skip is not None else 1
        self.num_radial = num_radial or 6
        self.num_spherical = num_spherical or 7
        self.radius = radius or 5.0
        self.is_edge_model = True
        super().__init__(edge_dim=edge_dim, **kwargs)
        self.rbf_layer = BesselBasisLayer(self.num_radial, self.radius,
                                          self.envelope_exponent)
        self.sbf_layer = SphericalBasisLayer(
            self.num_spherical, self.num_radial, self.radius,
            self.envelope_exponent)
        self.emb_block = HydraEmbeddingBlock(self.input_dim, self.hidden_dim,
                                             self.num_radial)

    def get_conv(self, input_dim, output_dim, edge_dim=None):
        return _DimeConv(self.hidden_dim, self.int_emb_size,
                         self.basis_emb_size, self.out_emb_size, output_dim,
                         self.num_radial, self.num_spherical,
                         self.num_before_skip, self.num_after_skip)

    def _embedding(self, data):
        pos = data.pos
        ei = data.edge_index
        vec, lengths = get_edge_vectors_and_lengths(
            pos, ei, data.get("edge_shifts"))
        dist = lengths.squeeze(-1)
        idx_kj, idx_ji = triplets(ei, data.num_nodes)
        # angle between v_ji (edge j->i) and v_jk = -v_kj (edge k->j)
        v_ji = vec[idx_ji]
        v_jk = -vec[idx_kj]
        dot = (v_ji * v_jk).sum(-1)
        cross = torch.linalg.norm(torch.cross(v_ji, v_jk, dim=-1), dim=-1)
        angle = torch.atan2(cross, dot)
        rbf = self.rbf_layer(dist)
        sbf = self.sbf_layer(dist, angle, idx_kj)
        x = data.x
        if not torch.is_floating_point(x):
            x = x.float()
        x = x.to(rbf.dtype)
        m = self.emb_block(x, rbf, ei)
        conv_args = {
            "edge_index": ei, "rbf": rbf, "sbf": sbf,
            "idx_kj": idx_kj, "idx_ji": idx_ji, "msg_state": {"m": m},
        }
        return x, pos, conv_args

    def __str__(self):
        return "DIMEStack"
