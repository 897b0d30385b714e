"""Property-based tests (hypothesis) for the aggregation ops' CPU
reference paths — the numerical ground truth the GPU kernels are
tested against, so these invariants transitively pin the kernels."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from hydragnn_amd.ops import gather, scatter, segment_softmax


@settings(max_examples=50, deadline=None)
@given(st.integers(1, 200), st.integers(1, 40), st.integers(1, 8),
       st.integers(0, 2 ** 31 - 1))
def test_scatter_sum_matches_index_add(E, N, F, seed):
    g = torch.Generator().manual_seed(seed)
    src = torch.randn(E, F, generator=g)
    idx = torch.randint(0, N, (E,), generator=g)
    out = scatter(src, idx, N, "sum")
    ref = torch.zeros(N, F).index_add_(0, idx, src)
    assert torch.allclose(out, ref, atol=1e-5)


@settings(max_examples=50, deadline=None)
@given(st.integers(1, 200), st.integers(1, 40), st.integers(0, 2**31 - 1))
def test_gather_scatter_adjoint(E, N, seed):
    """<gather(x), y> == <x, scatter(y)>: gather and scatter-sum are
    adjoint linear maps (this IS the autograd closure)."""
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(N, 4, generator=g)
    y = torch.randn(E, 4, generator=g)
    idx = torch.randint(0, N, (E,), generator=g)
    lhs = (gather(x, idx) * y).sum()
    rhs = (x * scatter(y, idx, N, "sum")).sum()
    assert torch.allclose(lhs, rhs, atol=1e-4)


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 150), st.integers(1, 30), st.integers(0, 2**31 - 1))
def test_segment_softmax_normalized(E, N, seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(E, generator=g)
    idx = torch.randint(0, N, (E,), generator=g)
    s = segment_softmax(x, idx, N)
    sums = scatter(s, idx, N, "sum")
    present = torch.bincount(idx, minlength=N) > 0
    assert torch.allclose(sums[present],
                          torch.ones(int(present.sum())), atol=1e-5)
    assert (s >= 0).all() and (s <= 1 + 1e-6).all()


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 100), st.integers(1, 20), st.integers(0, 2**31 - 1))
def test_scatter_mean_bounded_by_extremes(E, N, seed):
    g = torch.Generator().manual_seed(seed)
    src = torch.randn(E, 3, generator=g)
    idx = torch.randint(0, N, (E,), generator=g)
    mean = scatter(src, idx, N, "mean")
    mx = scatter(src, idx, N, "max")
    mn = scatter(src, idx, N, "min")
    present = torch.bincount(idx, minlength=N) > 0
    assert (mean[present] <= mx[present] + 1e-5).all()
    assert (mean[present] >= mn[present] - 1e-5).all()


@settings(max_examples=20, deadline=None)
@given(st.integers(2, 60), st.integers(0, 2**31 - 1))
def test_radius_graph_symmetric_no_loops(N, seed):
    from hydragnn_amd.ops import radius_graph
    g = torch.Generator().manual_seed(seed)
    pos = torch.rand(N, 3, generator=g)
    ei = radius_graph(pos, 0.5, max_num_neighbors=N)
    src, dst = ei[0], ei[1]
    assert (src != dst).all()  # no self loops by default
    d = (pos[src] - pos[dst]).norm(dim=-1)
    assert (d <= 0.5 + 1e-6).all()
    # symmetry: each (i, j) has (j, i) unless the neighbor cap bit
    pairs = {(int(a), int(b)) for a, b in ei.t().tolist()}
    sym = all((b, a) in pairs for (a, b) in pairs)
    assert sym
