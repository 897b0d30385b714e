"""hipGraph-captured train step vs eager on a real GPU: same data,
same init -> closely matching losses and parameters."""

import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                               reason="requires GPU")


def _run_epochs(capture: bool, steps=6, batch=16, precision="bf16"):
    from torch.utils.data import DataLoader

    import bench as bench_mod
    from hydragnn_amd.preprocess.static_batch import (
        StaticShapeCollater, compute_static_caps)
    from hydragnn_amd.train import train
    from hydragnn_amd.utils.datasets.synthetic import (
        md17_shape_dataset_fast)

    os.environ["HYDRAGNN_CAPTURE"] = "1" if capture else "0"
    try:
        torch.manual_seed(23)
        model = bench_mod.build_model("cuda:0", precision=precision,
                                      seed=23)
        opt = torch.optim.AdamW(model.parameters(), lr=1e-3,
                                foreach=True)
        ds = md17_shape_dataset_fast(steps * batch, seed=77)
        nc, ec = compute_static_caps(ds, batch)
        loader = DataLoader(ds, batch_size=batch, shuffle=False,
                            collate_fn=StaticShapeCollater(nc, ec, 28.0))
        losses = []
        for _ in range(2):
            err, tasks = train(loader, model, opt, 0,
                               precision=precision)
            losses.append(float(err))
        params = torch.cat([p.detach().float().flatten()
                            for p in model.parameters()]).cpu()
        if capture:
            base = model.module if hasattr(model, "module") else model
            stepper = getattr(base, "_hip_captured_step", None)
            assert stepper not in (None, False), \
                "captured path did not engage"
            base._hip_captured_step = None
        return losses, params
    finally:
        os.environ.pop("HYDRAGNN_CAPTURE", None)
        import gc
        gc.collect()
        torch.cuda.empty_cache()


@needs_gpu
def test_captured_train_matches_eager():
    l_eager, p_eager = _run_epochs(capture=False)
    l_cap, p_cap = _run_epochs(capture=True)
    for a, b in zip(l_eager, l_cap):
        assert abs(a - b) / max(1.0, abs(a)) < 0.05, (l_eager, l_cap)
    diff = (p_eager - p_cap).abs().max()
    scale = p_eager.abs().max()
    assert diff < 0.02 * max(1.0, float(scale)), float(diff)


@needs_gpu
def test_captured_second_epoch_reuses_graph():
    """Graph is captured once and cached on the model across train()
    calls (the bench warmup/timed epochs share it)."""
    from torch.utils.data import DataLoader

    import bench as bench_mod
    from hydragnn_amd.preprocess.static_batch import (
        StaticShapeCollater, compute_static_caps)
    from hydragnn_amd.train import train
    from hydragnn_amd.utils.datasets.synthetic import (
        md17_shape_dataset_fast)

    torch.manual_seed(5)
    model = bench_mod.build_model("cuda:0", precision="bf16_pure",
                                  seed=5)
    from hydragnn_amd.ops.fused_adamw import FusedAdamW
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    ds = md17_shape_dataset_fast(32, seed=3)
    nc, ec = compute_static_caps(ds, 16)
    loader = DataLoader(ds, batch_size=16, shuffle=False,
                        collate_fn=StaticShapeCollater(nc, ec, 28.0))
    train(loader, model, opt, 0, precision="bf16_pure")
    s1 = getattr(model, "_hip_captured_step", None)
    assert s1 not in (None, False)
    train(loader, model, opt, 0, precision="bf16_pure")
    s2 = getattr(model, "_hip_captured_step", None)
    assert s2 is s1, "graph was re-captured"
    model._hip_captured_step = None
    del s1, s2
    import gc
    gc.collect()
    torch.cuda.empty_cache()


@needs_gpu
def test_config_driven_capture_engages():
    """Training.Batching.mode='static_shape' through create_dataloaders
    engages the hipGraph-captured step on GPU."""
    from hydragnn_amd.preprocess import create_dataloaders
    from hydragnn_amd.train import train
    from hydragnn_amd.utils.datasets.synthetic import (
        md17_shape_dataset_fast)
    import bench as bench_mod

    torch.manual_seed(9)
    model = bench_mod.build_model("cuda:0", precision="bf16_pure",
                                  seed=9)
    from hydragnn_amd.ops.fused_adamw import FusedAdamW
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    ds = md17_shape_dataset_fast(48, seed=21)
    config = {"NeuralNetwork": {"Training": {
        "batch_size": 16,
        "Batching": {"mode": "static_shape", "pad_spacing": 28.0}}}}
    loaders = create_dataloaders(ds, ds, ds, 16, config=config)
    err, _ = train(loaders[0], model, opt, 0, precision="bf16_pure")
    assert torch.isfinite(err).all()
    stepper = getattr(model, "_hip_captured_step", None)
    assert stepper not in (None, False), "capture did not engage"
    model._hip_captured_step = None
    import gc
    gc.collect()
    torch.cuda.empty_cache()


@needs_gpu
def test_captured_mixed_shapes_falls_back_eager():
    """A loader mixing static-collated and plain batches: the stepper
    engages on matching shapes and the mismatched batch runs eager —
    both finite, no crash."""
    from torch.utils.data import DataLoader

    import bench as bench_mod
    from hydragnn_amd.data import Batch
    from hydragnn_amd.preprocess.static_batch import (
        StaticShapeCollater, compute_static_caps)
    from hydragnn_amd.train import train
    from hydragnn_amd.utils.datasets.synthetic import (
        md17_shape_dataset_fast)

    torch.manual_seed(13)
    model = bench_mod.build_model("cuda:0", precision="bf16_pure",
                                  seed=13)
    from hydragnn_amd.ops.fused_adamw import FusedAdamW
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    ds = md17_shape_dataset_fast(48, seed=31)
    nc, ec = compute_static_caps(ds, 16)
    coll = StaticShapeCollater(nc, ec, 28.0)

    class MixedLoader:
        def __iter__(self):
            yield coll(ds[:16])                       # static
            yield Batch.from_data_list(
                [d.clone() for d in ds[16:28]])       # 12 graphs, raw
            yield coll(ds[32:48])                     # static again

        def __len__(self):
            return 3

    err, _ = train(MixedLoader(), model, opt, 0, precision="bf16_pure")
    assert torch.isfinite(err).all()
    stepper = getattr(model, "_hip_captured_step", None)
    assert stepper not in (None, False)
    model._hip_captured_step = None
    import gc
    gc.collect()
    torch.cuda.empty_cache()
