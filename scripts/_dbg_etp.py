import sys, torch
sys.path.insert(0, "tests"); sys.path.insert(0, ".")
from hydragnn_amd.ops.etp import ETPMeta, _etp_indexed_dense, etp_indexed
from hydragnn_amd.ops.scatter import _rowptr_from_sorted
from test_gpu_etp import _table

g = torch.Generator().manual_seed(1)
N, E, C, da, db, dg, do = 60, 900, 8, 4, 9, 6, 16
tab = _table(1, da=da, db=db, dg=dg, do=do)
src = torch.randint(0, N, (E,), generator=g).cuda()
dst = torch.randint(0, N, (E,), generator=g).cuda()
eid_d = torch.argsort(dst, stable=True)
rowptr = _rowptr_from_sorted(dst[eid_d], N)
meta = ETPMeta(E, ai=src[eid_d], bi=eid_d, ci=eid_d, rowptr=rowptr, n_a_rows=N)
A0 = torch.randn(N, C, da, device="cuda")
B0 = torch.randn(E, db, device="cuda")
C0 = torch.randn(E, C, dg, device="cuda")

def run(dense, which):
    A = A0.clone().requires_grad_(True)
    B = B0.clone().requires_grad_(True)
    Cw = C0.clone().requires_grad_(True)
    fn = _etp_indexed_dense if dense else etp_indexed
    out = fn(A, B, Cw, tab, meta)
    loss = (out.float() ** 2).sum()
    gA, gB, gC = torch.autograd.grad(loss, (A, B, Cw), create_graph=True)
    term = {"A": gA, "B": gB, "C": gC}[which]
    loss2 = (term ** 2).sum()
    return torch.autograd.grad(loss2, (A, B, Cw), allow_unused=True)

for which in ("A", "B", "C"):
    f = run(False, which)
    d = run(True, which)
    for nm, a, b in zip("ABC", f, d):
        if a is None and b is None:
            continue
        if a is None or b is None:
            print(which, nm, "NONE mismatch", a is None, b is None)
            continue
        rel = (a - b).abs().max() / b.abs().max().clamp(min=1e-6)
        print(f"|g{which}|^2 -> d{nm}: rel {float(rel):.2e}")
