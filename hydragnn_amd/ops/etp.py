"""Generalized equivariant tensor-product contraction (ETP).

The primitive:  out[i, c, o] = sum_k coef_k A[i,c,a_k] B[i,b_k] C[i,c,g_k]
with a small constant entry table — covers the MACE edge tensor product
(A = gathered node feats, B = spherical harmonics, C = radial-MLP path
weights), its gradients, and the symmetric-contraction fold steps.

Key property: gradients of a trilinear form are trilinear forms with
role-permuted tables, and the channel-reduced B-gradient closes the
family.  So both autograd passes of force training (create_graph=True)
run on the two fused HIP kernels (csrc/etp.hip); CPU and fp64 use a
dense-einsum path that autograd differentiates directly.
"""

from __future__ import annotations

from typing import Dict, Tuple

import torch

from ._extension import get_extension, use_eager
from .scatter import _rowptr_from_sorted

_KERNEL_DTYPES = (torch.float32, torch.bfloat16, torch.float16,
                  torch.float64)


class ETPTable:
    """Entry table (a, b, g, o, coef) with dims (da, db, dg, do).
    Role permutations and device copies are cached."""

    def __init__(self, entries: torch.Tensor, coefs: torch.Tensor,
                 dims: Tuple[int, int, int, int]):
        assert entries.dim() == 2 and entries.shape[1] == 4
        self.entries = entries.to(torch.int64).cpu()
        self.coefs = coefs.to(torch.float32).cpu()
        self.dims = tuple(int(d) for d in dims)
        self._perm_cache: Dict[str, "ETPTable"] = {}
        self._dev_cache: Dict[Tuple, Tuple] = {}
        self._dense_cache: Dict[Tuple, torch.Tensor] = {}

    def perm(self, order: str) -> "ETPTable":
        """order: 4-char permutation of 'abgo' giving the new roles,
        e.g. 'obga' swaps A <-> out (the A-gradient table)."""
        if order in self._perm_cache:
            return self._perm_cache[order]
        col = {"a": 0, "b": 1, "g": 2, "o": 3}
        idx = [col[ch] for ch in order]
        ent = self.entries[:, idx].contiguous()
        dims = tuple(self.dims[i] for i in idx)
        t = ETPTable(ent, self.coefs, dims)
        self._perm_cache[order] = t
        return t

    def device_tensors(self, device):
        key = (device,)
        if key in self._dev_cache:
            return self._dev_cache[key]
        o = self.entries[:, 3]
        order = torch.argsort(o, stable=True)
        ent = self.entries[order].to(torch.int32)
        coefs = self.coefs[order]
        do = self.dims[3]
        counts = torch.bincount(o[order], minlength=do)
        starts = torch.zeros(do, dtype=torch.long)
        starts[1:] = counts.cumsum(0)[:-1]
        o_ranges = torch.stack([starts, counts], dim=1).to(torch.int32)
        out = (ent.to(device).contiguous(),
               coefs.to(device).contiguous(),
               o_ranges.to(device).contiguous())
        self._dev_cache[key] = out
        return out

    def dense(self, device, dtype) -> torch.Tensor:
        key = (device, dtype)
        if key in self._dense_cache:
            return self._dense_cache[key]
        da, db, dg, do = self.dims
        W = torch.zeros(da, db, dg, do, dtype=dtype, device=device)
        e = self.entries
        W[e[:, 0], e[:, 1], e[:, 2], e[:, 3]] = \
            self.coefs.to(dtype).to(device)
        self._dense_cache[key] = W
        return W


def _dense_general(A, B, C, table: ETPTable):
    W = table.dense(A.device, torch.float32).to(A.dtype)
    return torch.einsum("eca,eb,ecg,abgo->eco", A, B, C, W)


def _kernel_ok(table: ETPTable, *tensors) -> bool:
    da, db, dg, do = table.dims
    # real constraint is the per-block LDS budget (256 thread slices +
    # staged entry table) — mirrors the check in csrc/etp.hip.  Kept
    # at the FULL-staging footprint even though etp_general can fall
    # to its L1-operand variant: the gradient kernels (etp_reduce)
    # still stage fully, and this predicate gates the whole family.
    # fp64 slices are 8 bytes (full-precision accumulation in LDS).
    t = tensors[0]
    acc = 8 if t.dtype == torch.float64 else 4
    lds_bytes = 256 * (da + db + dg + do + 1) * acc + \
        table.entries.shape[0] * 20
    if lds_bytes > 150 * 1024 or max(da, db, dg, do) > 192:
        return False
    return (t.is_cuda and t.dtype in _KERNEL_DTYPES
            and all(x.dtype == t.dtype for x in tensors
                    if torch.is_tensor(x))
            and not use_eager())


class _ETPGeneral(torch.autograd.Function):
    @staticmethod
    def forward(ctx, A, B, C, table):
        ctx.save_for_backward(A, B, C)
        ctx.table = table
        ext = get_extension(required=True)
        ent, coefs, o_ranges = table.device_tensors(A.device)
        return ext.etp_general(A.contiguous(), B.contiguous(),
                               C.contiguous(), ent, coefs, o_ranges,
                               table.dims[3])

    @staticmethod
    def backward(ctx, gout):
        A, B, C = ctx.saved_tensors
        table = ctx.table
        gout = gout.contiguous()
        gA = gB = gC = None
        if ctx.needs_input_grad[0]:
            gA = etp_general(gout, B, C, table.perm("obga"))
        if ctx.needs_input_grad[1]:
            gB = etp_reduce(A, C, gout, table)
        if ctx.needs_input_grad[2]:
            gC = etp_general(A, B, gout, table.perm("abog"))
        return gA, gB, gC, None


class _ETPReduce(torch.autograd.Function):
    """out[e, b] = sum_c sum_k coef A[.,a] C[.,g] D[.,o]."""

    @staticmethod
    def forward(ctx, A, C, D, table):
        ctx.save_for_backward(A, C, D)
        ctx.table = table
        ext = get_extension(required=True)
        ent, coefs, _ = table.device_tensors(A.device)
        return ext.etp_reduce(A.contiguous(), C.contiguous(),
                              D.contiguous(), ent, coefs, table.dims[1])

    @staticmethod
    def backward(ctx, gout):
        A, C, D = ctx.saved_tensors
        table = ctx.table
        gout = gout.contiguous()
        gA = gC = gD = None
        if ctx.needs_input_grad[0]:
            # gA[e,c,a] = sum coef gout[b] C[g] D[o]
            gA = etp_general(C, gout, D, table.perm("gboa"))
        if ctx.needs_input_grad[1]:
            # gC[e,c,g] = sum coef A[a] gout[b] D[o]:
            # roles A=a, B=b, C=o, out=g
            gC = etp_general(A, gout, D, table.perm("abog"))
        if ctx.needs_input_grad[2]:
            # gD[e,c,o] = sum coef A[a] gout[b] C[g]: base role order
            gD = etp_general(A, gout, C, table.perm("abgo"))
        return gA, gC, gD, None


def etp_general(A: torch.Tensor, B: torch.Tensor, C: torch.Tensor,
                table: ETPTable) -> torch.Tensor:
    """out[i,c,o] = sum_k coef_k A[i,c,a] B[i,b] C[i,c,g]."""
    if _kernel_ok(table, A, B, C):
        return _ETPGeneral.apply(A, B, C, table)
    return _dense_general(A, B, C, table)


_FOLD_TABLES: Dict[Tuple[int, int], ETPTable] = {}


def fold_last(t: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    """out[n,c,P] = sum_i t[n,c,P,i] x[n,c,i] — the symmetric
    contraction fold step, via the fused ETP kernel when dims fit."""
    n, c = t.shape[0], t.shape[1]
    D = t.shape[-1]
    P = t.numel() // (n * c * D)
    key = (P, D)
    if key not in _FOLD_TABLES:
        ents = [(p * D + i, 0, i, p) for p in range(P) for i in range(D)]
        _FOLD_TABLES[key] = ETPTable(
            torch.tensor(ents, dtype=torch.long),
            torch.ones(len(ents)), (P * D, 1, D, P))
    table = _FOLD_TABLES[key]
    A = t.reshape(n, c, P * D)
    if _kernel_ok(table, A, x, x):
        ones = torch.ones(n, 1, dtype=t.dtype, device=t.device)
        out = etp_general(A, ones, x, table)
    else:
        out = torch.einsum("ncpi,nci->ncp", t.reshape(n, c, P, D), x)
    return out.reshape(t.shape[:-1])


def etp_reduce(A: torch.Tensor, C: torch.Tensor, D: torch.Tensor,
               table: ETPTable) -> torch.Tensor:
    """out[e,b] = sum_c sum_k coef_k A[e,c,a] C[e,c,g] D[e,c,o]."""
    if (table.dims[1] <= 12 and A.is_cuda
            and A.dtype in _KERNEL_DTYPES and not use_eager()):
        return _ETPReduce.apply(A, C, D, table)
    W = table.dense(A.device, torch.float32).to(A.dtype)
    return torch.einsum("eca,ecg,eco,abgo->eb", A, C, D, W)


# ---------------------------------------------------------------------------
# Indexed / CSR-fused family: gather + TP + segment-sum in one kernel.
#
# Primitive (positions e walk a chosen edge ordering):
#   OUT[r, c, o] = sum_{e in rowptr[r]..rowptr[r+1]} sum_k coef_k
#                  A[ai[e], c, a] B[bi[e], b] C[ci[e], c, g]
# Gradients are instances of the SAME primitive with re-sorted positions
# (see ETPMeta.grad_meta_a), so force training stays fused end to end.
# ---------------------------------------------------------------------------
class ETPMeta:
    """Edge-ordering metadata: per-slot row indices + optional output
    CSR.  bi/ci must be bijections (or None = identity); ai may be a
    many-to-one node map."""

    def __init__(self, n_positions, ai=None, bi=None, ci=None,
                 rowptr=None, n_a_rows=None):
        self.n_positions = int(n_positions)
        self.ai = ai
        self.bi = bi
        self.ci = ci
        self.rowptr = rowptr
        self.n_a_rows = n_a_rows  # rows of A (for the gA output)
        self._r_of_pos = None
        self._grad_meta_a = None

    def r_of_pos(self):
        """Output row of each position (identity without CSR)."""
        if self.rowptr is None:
            return None
        if self._r_of_pos is None:
            counts = self.rowptr[1:] - self.rowptr[:-1]
            self._r_of_pos = torch.repeat_interleave(
                torch.arange(counts.numel(),
                             device=self.rowptr.device), counts)
        return self._r_of_pos

    def grad_meta_a(self):
        """Meta for the A-slot gradient: positions re-sorted by ai so
        the output CSR accumulates over A's rows."""
        if self._grad_meta_a is None:
            assert self.ai is not None and self.n_a_rows is not None, \
                "A-slot gradient needs ai + n_a_rows"
            rho = torch.argsort(self.ai, stable=True)
            rowptr_a = _rowptr_from_sorted(self.ai[rho], self.n_a_rows)
            r = self.r_of_pos()
            ai_new = r[rho] if r is not None else rho
            bi_new = self.bi[rho] if self.bi is not None else rho
            ci_new = self.ci[rho] if self.ci is not None else rho
            # the gradient instance's A is the original OUT tensor:
            # its row count (for second-order CSR) is R (CSR mode) or
            # the position count (per-edge mode)
            n_rows_gA = (self.rowptr.numel() - 1
                         if self.rowptr is not None
                         else self.n_positions)
            m = ETPMeta(self.n_positions, ai=ai_new, bi=bi_new,
                        ci=ci_new, rowptr=rowptr_a,
                        n_a_rows=n_rows_gA)
            self._grad_meta_a = (rho, m)
        return self._grad_meta_a


def _pos_index(t, idx, E):
    if idx is None:
        return t[:E]
    return t.index_select(0, idx)


def _etp_indexed_dense(A, B, C, table, meta):
    """Differentiable fallback (CPU / fp64): gather positions, dense
    einsum, CSR accumulate."""
    E = meta.n_positions
    Apos = _pos_index(A, meta.ai, E)
    Bpos = _pos_index(B, meta.bi, E)
    Cpos = _pos_index(C, meta.ci, E)
    out_pos = _dense_general(Apos, Bpos, Cpos, table)
    if meta.rowptr is None:
        return out_pos
    r = meta.r_of_pos()
    R = meta.rowptr.numel() - 1
    out = out_pos.new_zeros(R, out_pos.shape[1], out_pos.shape[2])
    out.index_add_(0, r, out_pos)
    return out


class _ETPIndexed(torch.autograd.Function):
    @staticmethod
    def forward(ctx, A, B, C, table, meta):
        ctx.save_for_backward(A, B, C)
        ctx.table = table
        ctx.meta = meta
        ext = get_extension(required=True)
        ent, coefs, o_ranges = table.device_tensors(A.device)
        if meta.rowptr is not None:
            return ext.etp_nodesum(A.contiguous(), B.contiguous(),
                                   C.contiguous(), ent, coefs,
                                   table.dims[3], meta.rowptr,
                                   meta.ai, meta.bi, meta.ci)
        return ext.etp_general(A.contiguous(), B.contiguous(),
                               C.contiguous(), ent, coefs, o_ranges,
                               table.dims[3], meta.ai, meta.bi,
                               meta.ci, meta.n_positions)

    @staticmethod
    def backward(ctx, gout):
        A, B, C = ctx.saved_tensors
        table = ctx.table
        meta = ctx.meta
        gout = gout.contiguous()
        E = meta.n_positions
        gA = gB = gC = None
        if ctx.needs_input_grad[0]:
            _, meta_a = meta.grad_meta_a()
            gA = etp_indexed(gout, B, C, table.perm("obga"), meta_a)
        if ctx.needs_input_grad[1]:
            # per-position reduce then un-permute via bi (bijection)
            gB_pos = _etp_reduce_idx(A, C, gout, table, meta)
            if meta.bi is not None:
                # index_add_: derived metas can carry many-to-one maps
                gB = gB_pos.new_zeros(B.shape[0], gB_pos.shape[1])
                gB = gB.index_add(0, meta.bi, gB_pos)
            else:
                gB = gB_pos
        if ctx.needs_input_grad[2]:
            r = meta.r_of_pos()
            meta_c = ETPMeta(E, ai=meta.ai, bi=meta.bi,
                             ci=r if r is not None else None,
                             rowptr=None, n_a_rows=A.shape[0])
            gC_pos = etp_indexed(A, B, gout, table.perm("abog"), meta_c)
            if meta.ci is not None:
                gC = gC_pos.new_zeros(C.shape)
                gC = gC.index_add(0, meta.ci, gC_pos)
            else:
                gC = gC_pos
        return gA, gB, gC, None, None


class _ETPReduceIdx(torch.autograd.Function):
    """Per-position channel-reduced contraction with row indices:
    out[e, b] = sum_c sum_k coef A[ai,c,a] C[ci,c,g] D[di,c,o]."""

    @staticmethod
    def forward(ctx, A, C, D, table, ai, ci, di, E):
        ctx.save_for_backward(A, C, D)
        ctx.table = table
        ctx.idx = (ai, ci, di)
        ctx.E = E
        ext = get_extension(required=True)
        ent, coefs, _ = table.device_tensors(A.device)
        return ext.etp_reduce(A.contiguous(), C.contiguous(),
                              D.contiguous(), ent, coefs,
                              table.dims[1], ai, ci, di, E)

    @staticmethod
    def backward(ctx, gout):
        A, C, D = ctx.saved_tensors
        ai, ci, di = ctx.idx
        E = ctx.E
        table = ctx.table
        gout = gout.contiguous()

        def scatter_rows(per_pos, idx, rows):
            if idx is None:
                return per_pos
            out = per_pos.new_zeros((rows,) + per_pos.shape[1:])
            # idx may be many-to-one (node map): accumulate
            return out.index_add_(
                0, idx, per_pos) if idx.numel() == per_pos.shape[0] \
                else out
        gA = gC = gD = None
        if ctx.needs_input_grad[0]:
            # gA[ai,c,a] += coef gout[b] C[g] D[o]
            m = ETPMeta(E, ai=ci, bi=None, ci=di, rowptr=None,
                        n_a_rows=C.shape[0])
            per = etp_indexed(C, gout, D, table.perm("gboa"), m)
            gA = scatter_rows(per, ai, A.shape[0]) if ai is not None \
                else per
        if ctx.needs_input_grad[1]:
            m = ETPMeta(E, ai=ai, bi=None, ci=di, rowptr=None,
                        n_a_rows=A.shape[0])
            per = etp_indexed(A, gout, D, table.perm("abog"), m)
            gC = scatter_rows(per, ci, C.shape[0]) if ci is not None \
                else per
        if ctx.needs_input_grad[2]:
            m = ETPMeta(E, ai=ai, bi=None, ci=ci, rowptr=None,
                        n_a_rows=A.shape[0])
            per = etp_indexed(A, gout, C, table.perm("abgo"), m)
            gD = scatter_rows(per, di, D.shape[0]) if di is not None \
                else per
        return gA, gC, gD, None, None, None, None, None


def _etp_reduce_idx(A, C, D, table, meta):
    r = meta.r_of_pos()
    di = r if r is not None else None
    if (A.is_cuda and A.dtype in _KERNEL_DTYPES and not use_eager()
            and table.dims[1] <= 12):
        return _ETPReduceIdx.apply(A, C, D, table, meta.ai, meta.ci,
                                   di, meta.n_positions)
    E = meta.n_positions
    Apos = _pos_index(A, meta.ai, E)
    Cpos = _pos_index(C, meta.ci, E)
    Dpos = D if di is None else D.index_select(0, di)
    W = table.dense(A.device, torch.float32).to(A.dtype)
    return torch.einsum("eca,ecg,eco,abgo->eb", Apos, Cpos, Dpos, W)


def etp_indexed(A, B, C, table, meta: ETPMeta) -> torch.Tensor:
    """Indexed / CSR-fused contraction (see ETPMeta)."""
    if _kernel_ok(table, A, B, C):
        return _ETPIndexed.apply(A, B, C, table, meta)
    return _etp_indexed_dense(A, B, C, table, meta)
