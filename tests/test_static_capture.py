"""Static-shape padding, weighted loss, fp32 grad sync, and the
train-loop bench path (CPU; the hipGraph capture itself is covered by
tests/test_gpu_capture.py)."""

import json
import os
import subprocess
import sys

import pytest
import torch
import torch.distributed as dist

from hydragnn_amd.data import Batch
from hydragnn_amd.preprocess.static_batch import (
    StaticShapeCollater, compute_static_caps, pad_batch_static)
from hydragnn_amd.utils.datasets.synthetic import (
    md17_shape_dataset, md17_shape_dataset_fast)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def _small_mace(dataset):
    from test_mace_model import _build, _mace_config
    model, config, _ = _build(_mace_config(), dataset)
    return model


def _loss_and_grads(model, batch):
    model.zero_grad()
    batch.pos.requires_grad_(True)
    pred = model(batch)
    loss, tasks = model.energy_force_loss(pred, batch, create_graph=True)
    loss.backward()
    g = torch.cat([p.grad.flatten() for p in model.parameters()
                   if p.grad is not None])
    return loss, tasks, g


def test_pad_batch_loss_grad_equivalence():
    """Padded batch (sentinel pad graph, loss_weight_g=0) gives the
    same loss and parameter gradients as the unpadded batch."""
    torch.manual_seed(3)
    ds = md17_shape_dataset(num_samples=4)
    model = _small_mace(ds)
    b1 = Batch.from_data_list([d.clone() for d in ds])
    b2 = Batch.from_data_list([d.clone() for d in ds])
    nc, ec = compute_static_caps(ds, 4)
    b2 = pad_batch_static(b2, nc + 5, ec + 17, pad_spacing=30.0)

    l1, t1, g1 = _loss_and_grads(model, b1)
    l2, t2, g2 = _loss_and_grads(model, b2)
    assert torch.isfinite(l2)
    assert abs(float(l1) - float(l2)) < 1e-5 * max(1.0, abs(float(l1)))
    for a, b in zip(t1, t2):
        assert abs(float(a) - float(b)) < 1e-4 * max(1.0, abs(float(a)))
    assert (g1 - g2).abs().max() < 1e-5 * max(1.0, g1.abs().max())


def test_static_collater_fixed_shapes():
    ds = md17_shape_dataset_fast(32, seed=9)
    nc, ec = compute_static_caps(ds, 8)
    coll = StaticShapeCollater(nc, ec, pad_spacing=28.0)
    shapes = None
    for i in range(0, 32, 8):
        b = coll(ds[i:i + 8])
        s = {k: tuple(v.shape) for k, v in b.items()
             if torch.is_tensor(v)}
        if shapes is None:
            shapes = s
        assert s == shapes
        assert b.num_nodes == nc and b.num_edges == ec
        assert bool(b.get("static_shape_"))
        assert int(b.get("num_real_graphs_")) == 8
        w = b["loss_weight_g"]
        assert w.shape == (9,) and float(w[-1]) == 0.0 \
            and float(w[:-1].sum()) == 8.0
        dst = b.edge_index[1]
        assert (dst[1:] >= dst[:-1]).all(), "padding broke dst sort"
        # pad edges never touch real nodes
        n_real = sum(d.num_nodes for d in ds[i:i + 8])
        pad_e = b.edge_index[:, -(ec - sum(d.num_edges
                                           for d in ds[i:i + 8])):]
        if pad_e.numel():
            assert int(pad_e.min()) >= n_real


def test_weighted_loss_matches_unweighted():
    from hydragnn_amd.models.create import _make_weighted_loss
    torch.manual_seed(0)
    pred = torch.randn(12, 3)
    true = torch.randn(12, 3)
    w = torch.ones(12)
    fn = _make_weighted_loss("mse", w)
    ref = torch.nn.functional.mse_loss(pred, true)
    assert torch.allclose(fn(pred, true), ref, atol=1e-6)
    fn = _make_weighted_loss("mae", w)
    ref = torch.nn.functional.l1_loss(pred, true)
    assert torch.allclose(fn(pred, true), ref, atol=1e-6)
    # zero-weight rows drop out
    w2 = torch.cat([torch.ones(8), torch.zeros(4)])
    fn = _make_weighted_loss("mse", w2)
    ref = torch.nn.functional.mse_loss(pred[:8], true[:8])
    assert torch.allclose(fn(pred, true), ref, atol=1e-6)


def test_train_loop_with_padded_batches():
    """train() (eager path on CPU) over a static-collated loader:
    finite losses, sample counting excludes the pad graph."""
    from torch.utils.data import DataLoader

    from hydragnn_amd.train import train

    torch.manual_seed(5)
    ds = md17_shape_dataset(num_samples=8)
    model = _small_mace(ds)
    nc, ec = compute_static_caps(ds, 4)
    loader = DataLoader(ds, batch_size=4,
                        collate_fn=StaticShapeCollater(nc, ec),
                        shuffle=False)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    err, tasks = train(loader, model, opt, 0)
    assert torch.isfinite(err).all()
    assert torch.isfinite(tasks).all()


def _grad_sync_worker(rank, world, port, q):
    try:
        os.environ.update(MASTER_ADDR="127.0.0.1",
                          MASTER_PORT=str(port), RANK=str(rank),
                          WORLD_SIZE=str(world), LOCAL_RANK=str(rank))
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from hydragnn_amd.parallel import FlatGradSync
        torch.manual_seed(100 + rank)
        m = torch.nn.Linear(8, 8).to(torch.bfloat16)
        x = torch.randn(4, 8).to(torch.bfloat16)
        m(x).float().pow(2).sum().backward()
        local = [p.grad.clone() for p in m.parameters()]
        sync = FlatGradSync(m.parameters())
        sync()
        # expected: fp32 mean across ranks, cast back to bf16
        for p, lg in zip(m.parameters(), local):
            gathered = [torch.zeros_like(lg, dtype=torch.float32)
                        for _ in range(world)]
            dist.all_gather(gathered, lg.float())
            want = (torch.stack(gathered).sum(0) / world).to(
                torch.bfloat16)
            assert torch.equal(p.grad, want), "fp32 mean mismatch"
        q.put((rank, True, ""))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))


def test_flat_grad_sync_gloo():
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_grad_sync_worker, args=(r, 2, 29661, q))
          for r in range(2)]
    for p in ps:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in ps:
        p.join(timeout=60)
    for rank, ok, info in results:
        assert ok, f"rank {rank}: {info}"


def test_bench_trainloop_cpu():
    """bench.py default path runs through train() on CPU and prints
    one valid JSON line."""
    env = dict(os.environ, HYDRAGNN_BENCH_PRECISION="fp32",
               HYDRAGNN_BENCH_WORKERS="0", MASTER_ADDR="127.0.0.1")
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--steps", "2",
         "--warmup", "1", "--batch", "4"],
        capture_output=True, text=True, timeout=420, cwd=REPO, env=env)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")][-1]
    out = json.loads(line)
    assert out["value"] > 0 and out["steps"] == 2
    assert out["n_gpus"] == 1


# (2-rank torchrun coverage of the same path lives in
# tests/test_distributed.py::test_bench_two_rank_gloo.)


def _capture_agreement_worker(rank, world, port, q):
    try:
        os.environ.update(MASTER_ADDR="127.0.0.1",
                          MASTER_PORT=str(port), RANK=str(rank),
                          WORLD_SIZE=str(world), LOCAL_RANK=str(rank))
        dist.init_process_group("gloo", rank=rank, world_size=world)
        import hydragnn_amd.train.captured as cap
        from hydragnn_amd.data import Data

        class _FakeStepper:
            def __init__(self, model, opt, batch, *a, **k):
                # rank 1 "fails" capture; rank 0 succeeds
                if dist.get_rank() == 1:
                    raise RuntimeError("no capture on this rank")
                self.opt = opt

            def matches(self, data):
                return True

        orig = cap.CapturedTrainStep
        cap.CapturedTrainStep = _FakeStepper
        try:
            model = torch.nn.Linear(2, 2)
            opt = torch.optim.SGD(model.parameters(), lr=0.1)
            batch = Data(pos=torch.zeros(3, 3),
                         x=torch.zeros(3, 2),
                         edge_index=torch.zeros(2, 0, dtype=torch.long))
            stepper = cap.get_or_build_stepper(
                model, opt, batch, None, None, torch.float32)
        finally:
            cap.CapturedTrainStep = orig
        # one rank failed -> ALL ranks must fall back to eager
        q.put((rank, stepper is None, ""))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))


def test_capture_rank_agreement_falls_back_everywhere():
    """If capture fails on any rank, every rank must run eager —
    a mixed fleet would mismatch collectives and hang the 8-GPU
    scaling run."""
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_capture_agreement_worker,
                      args=(r, 2, 29681, q)) for r in range(2)]
    for p in ps:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in ps:
        p.join(timeout=60)
    for rank, fell_back, info in results:
        assert fell_back, f"rank {rank}: stepper not None ({info})"


def test_config_driven_static_shape_batching():
    """Training.Batching.mode == 'static_shape' produces fixed-shape
    padded train batches through create_dataloaders, and train() runs
    over them (the config path to the captured step)."""
    from hydragnn_amd.preprocess import create_dataloaders
    from hydragnn_amd.train import train

    torch.manual_seed(6)
    ds = md17_shape_dataset(num_samples=16)
    model = _small_mace(ds)
    config = {"NeuralNetwork": {"Training": {
        "batch_size": 4,
        "Batching": {"mode": "static_shape", "pad_spacing": 28.0}}}}
    tr_loader, val_loader, te_loader = create_dataloaders(
        ds, ds, ds, 4, config=config)
    shapes = None
    for b in tr_loader:
        assert bool(b.get("static_shape_"))
        s = {k: tuple(v.shape) for k, v in b.items()
             if torch.is_tensor(v)}
        shapes = shapes or s
        assert s == shapes
    # eval loaders stay exact (no pad graph)
    for b in val_loader:
        assert b.get("static_shape_") is None
        break
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    err, _ = train(tr_loader, model, opt, 0)
    assert torch.isfinite(err).all()


def test_capture_gate_requires_shape_static_model():
    """Models with data-dependent internal shapes (SchNet dynamic
    radius rebuild, DimeNet triplets, dense global attention) must NOT
    enter the captured path even with static-collated batches."""
    from hydragnn_amd.train.captured import capture_enabled
    from hydragnn_amd.models.create import create_model

    mace = _small_mace(md17_shape_dataset(num_samples=4))
    assert mace.supports_hipgraph_capture
    schnet = create_model(
        mpnn_type="SchNet", input_dim=1, hidden_dim=16,
        output_dim=[1], output_type=["graph"],
        output_heads={"graph": [{"type": "branch-0", "architecture": {
            "num_sharedlayers": 1, "dim_sharedlayers": 16,
            "num_headlayers": 1, "dim_headlayers": [16]}}]},
        activation_function="silu", loss_function_type="mse",
        task_weights=[1.0], num_conv_layers=1, radius=2.5,
        max_neighbours=10, num_gaussians=8, num_filters=16,
        use_gpu=False)
    assert not schnet.supports_hipgraph_capture
    gin_gps = create_model(
        mpnn_type="GIN", input_dim=1, hidden_dim=16,
        output_dim=[1], output_type=["graph"],
        output_heads={"graph": [{"type": "branch-0", "architecture": {
            "num_sharedlayers": 1, "dim_sharedlayers": 16,
            "num_headlayers": 1, "dim_headlayers": [16]}}]},
        activation_function="silu", loss_function_type="mse",
        task_weights=[1.0], num_conv_layers=1,
        global_attn_engine="gps", global_attn_type="multihead",
        global_attn_heads=2, pe_dim=2, use_gpu=False)
    assert not gin_gps.supports_hipgraph_capture


def test_config_driven_bf16_pure_precision():
    """Training.precision: bf16_pure casts the model to bf16 (no
    autocast layer) through the standard config flow."""
    from hydragnn_amd.models.create import resolve_precision
    prec, pd, ac = resolve_precision("bf16_pure")
    assert prec == "bf16_pure" and pd == torch.bfloat16 and ac is None
    prec2, pd2, _ = resolve_precision("bf16-pure")
    assert prec2 == "bf16_pure" and pd2 == torch.bfloat16

    from torch.utils.data import DataLoader
    from hydragnn_amd.preprocess.load_data import _collate
    from hydragnn_amd.train import train

    ds = md17_shape_dataset(num_samples=8)
    model = _small_mace(ds).to(torch.bfloat16)
    loader = DataLoader(ds, batch_size=4, collate_fn=_collate)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    err, _ = train(loader, model, opt, 0, precision="bf16_pure")
    assert torch.isfinite(err.float()).all()


@pytest.mark.parametrize("n_samples,extra_nodes,extra_edges", [
    (1, 0, 0),      # single-graph batch, minimum pad (2 nodes, 0 edges)
    (1, 3, 1),      # pad edges on a tiny pad graph
    (2, 0, 64),     # pad edges without extra nodes beyond the margin
    (3, 17, 0),     # many pad nodes, zero pad edges
])
def test_pad_batch_edge_cases(n_samples, extra_nodes, extra_edges):
    """Padding equivalence holds at the boundary shapes: minimum pad
    graph (2 nodes), zero pad edges, single-sample batches."""
    torch.manual_seed(11)
    ds = md17_shape_dataset(num_samples=n_samples)
    model = _small_mace(ds)
    b1 = Batch.from_data_list([d.clone() for d in ds])
    b2 = Batch.from_data_list([d.clone() for d in ds])
    nc = b1.num_nodes + 2 + extra_nodes
    ec = b1.num_edges + extra_edges
    b2 = pad_batch_static(b2, nc, ec, pad_spacing=30.0)
    assert b2.num_nodes == nc and b2.num_edges == ec
    l1, t1, g1 = _loss_and_grads(model, b1)
    l2, t2, g2 = _loss_and_grads(model, b2)
    assert torch.isfinite(l2)
    assert abs(float(l1) - float(l2)) < 1e-5 * max(1.0, abs(float(l1)))
    assert (g1 - g2).abs().max() < 1e-5 * max(1.0, g1.abs().max())


def test_pad_batch_rejects_impossible_caps():
    ds = md17_shape_dataset(num_samples=2)
    b = Batch.from_data_list(list(ds))
    with pytest.raises(ValueError, match="node_cap"):
        pad_batch_static(Batch.from_data_list(list(ds)),
                         b.num_nodes + 1, b.num_edges + 8)
    with pytest.raises(ValueError, match="edge_cap"):
        pad_batch_static(Batch.from_data_list(list(ds)),
                         b.num_nodes + 8, b.num_edges - 1)


@pytest.mark.parametrize("loss_type,ref_fn", [
    ("mse", torch.nn.MSELoss()),
    ("mae", torch.nn.L1Loss()),
    ("smooth_l1", torch.nn.SmoothL1Loss()),
    ("huber", torch.nn.HuberLoss()),
])
def test_weighted_loss_all_types(loss_type, ref_fn):
    """All-ones weights reproduce the torch loss; zero-weight rows are
    excluded exactly (pads contribute nothing for any loss type)."""
    from hydragnn_amd.models.create import _make_weighted_loss
    torch.manual_seed(5)
    pred = torch.randn(6, 3)
    true = torch.randn(6, 3)
    ones = _make_weighted_loss(loss_type, torch.ones(6))
    assert torch.allclose(ones(pred, true), ref_fn(pred, true),
                          atol=1e-6)
    w = torch.tensor([1.0, 1.0, 0.0, 1.0, 0.0, 1.0])
    masked = _make_weighted_loss(loss_type, w)
    keep = w.bool()
    assert torch.allclose(masked(pred, true),
                          ref_fn(pred[keep], true[keep]), atol=1e-6)


def test_weighted_loss_rmse():
    from hydragnn_amd.models.create import _make_weighted_loss
    torch.manual_seed(6)
    pred, true = torch.randn(5, 2), torch.randn(5, 2)
    fn = _make_weighted_loss("rmse", torch.ones(5))
    ref = torch.sqrt(torch.nn.functional.mse_loss(pred, true) + 1e-12)
    assert torch.allclose(fn(pred, true), ref, atol=1e-6)


def test_pnaplus_pad_equivalence():
    """PNAPlus is capture-safe: the Bessel envelope zeroes pad edges
    (all longer than cutoff), so padded batches match unpadded ones."""
    import sys
    sys.path.insert(0, os.path.dirname(__file__))
    from deterministic_graph_data import base_config, make_deterministic_dataset
    from hydragnn_amd.models import create_model_config
    from hydragnn_amd.preprocess import create_dataloaders
    from hydragnn_amd.utils.config import update_config

    torch.manual_seed(2)
    cfg = base_config("PNAPlus", heads=("graph",))
    ds = make_deterministic_dataset(num_samples=4, num_heads_node=0,
                                    include_graph_head=True)
    loaders = create_dataloaders(ds, ds, ds, 4, config=cfg)
    cfg = update_config(cfg, *loaders)
    model = create_model_config(cfg["NeuralNetwork"], use_gpu=False)
    assert model.supports_hipgraph_capture

    def strip(d):
        d = d.clone()
        if "y_loc" in d.keys():
            delattr(d, "y_loc")
        return d

    b1 = Batch.from_data_list([strip(d) for d in ds])
    b2 = Batch.from_data_list([strip(d) for d in ds])
    nc, ec = compute_static_caps(ds, 4)
    b2 = pad_batch_static(b2, nc + 4, ec + 9, pad_spacing=30.0)
    model.eval()
    with torch.no_grad():
        p1 = model(b1)
        p2 = model(b2)
    # real-graph predictions bitwise unaffected by padding
    assert torch.allclose(p1[0], p2[0][: p1[0].shape[0]], atol=1e-6)


def test_multibranch_disables_capture():
    """Multi-branch (multidataset) models partition by dataset_name
    with data-dependent masks — the capture gate must exclude them."""
    import sys
    sys.path.insert(0, os.path.dirname(__file__))
    from deterministic_graph_data import base_config, make_deterministic_dataset
    from hydragnn_amd.models import create_model_config
    from hydragnn_amd.preprocess import create_dataloaders
    from hydragnn_amd.utils.config import update_config

    cfg = base_config("GIN", heads=("graph",))
    heads = cfg["NeuralNetwork"]["Architecture"]["output_heads"]
    base_branch = heads["graph"]
    # two branches -> multidataset-style branched heads
    if isinstance(base_branch, dict):
        heads["graph"] = [
            {"type": f"branch-{i}", "architecture": dict(base_branch)}
            for i in range(2)]
    ds = make_deterministic_dataset(num_samples=4, num_heads_node=0,
                                    include_graph_head=True)
    loaders = create_dataloaders(ds, ds, ds, 2, config=cfg)
    cfg = update_config(cfg, *loaders)
    model = create_model_config(cfg["NeuralNetwork"], use_gpu=False)
    assert model.num_branches == 2
    assert not model.supports_hipgraph_capture


def test_conv_checkpointing_disables_capture():
    ds = md17_shape_dataset(num_samples=2)
    model = _small_mace(ds)
    core = model
    assert core.supports_hipgraph_capture
    core.enable_conv_checkpointing()
    assert not core.supports_hipgraph_capture


def test_capture_safety_marker_matrix():
    """Class-level capture-safety markers: every shape-static stack
    opts in; dynamic-shape stacks stay out."""
    from hydragnn_amd.models.egnn import EGCLStack
    from hydragnn_amd.models.mace.stack import MACEStack
    from hydragnn_amd.models.painn import PAINNStack
    from hydragnn_amd.models.pna_plus import PNAPlusStack
    from hydragnn_amd.models.pnaeq import PNAEqStack
    from hydragnn_amd.models.stacks import (CGCNNStack, GATStack,
                                            GINStack, MFCStack, PNAStack,
                                            SAGEStack)
    from hydragnn_amd.models.schnet import SCFStack
    from hydragnn_amd.models.dimenet import DIMEStack

    safe = (MACEStack, EGCLStack, PAINNStack, PNAEqStack, PNAPlusStack,
            GINStack, SAGEStack, MFCStack, CGCNNStack, PNAStack,
            GATStack)
    unsafe = (SCFStack, DIMEStack)
    for cls in safe:
        assert cls._hipgraph_capture_safe, cls.__name__
    for cls in unsafe:
        assert not cls._hipgraph_capture_safe, cls.__name__
