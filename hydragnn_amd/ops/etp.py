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

_KERNEL_DTYPES = (torch.float32, torch.bfloat16, torch.float16)


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
    # staged entry table) — mirrors the check in csrc/etp.hip
    lds_bytes = 256 * (da + db + dg + do + 1) * 4 + \
        table.entries.shape[0] * 20
    if lds_bytes > 150 * 1024 or max(da, db, dg, do) > 192:
        return False
    t = tensors[0]
    return (t.is_cuda and t.dtype in _KERNEL_DTYPES and not use_eager())


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
