"""Multi-rank behavior tested without a cluster: 2 spawned processes
with the gloo backend (reference CI strategy)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

pytestmark = pytest.mark.mpi


def _init(rank, world_size, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["LOCAL_RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)


def _ddp_training_worker(rank, world_size, port, q):
    try:
        _init(rank, world_size, port)
        import sys
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        from _training_workflow import run_training
        torch.manual_seed(7)
        from hydragnn_amd.utils.distributed import distributed_model_wrapper
        from hydragnn_amd.models import create_model_config
        from hydragnn_amd.preprocess import create_dataloaders, split_dataset
        from hydragnn_amd.utils.config import update_config
        from hydragnn_amd.utils.optimizer import select_optimizer
        from hydragnn_amd.train import train as train_fn
        from deterministic_graph_data import (base_config,
                                              make_deterministic_dataset)
        config = base_config("GIN", heads=("graph",), num_epoch=3)
        dataset = make_deterministic_dataset(num_samples=32,
                                             num_heads_node=0)
        tr, va, te = split_dataset(dataset, 0.7, seed=0)
        loaders = create_dataloaders(tr, va, te, 8, config=config)
        config = update_config(config, *loaders)
        model = create_model_config(config["NeuralNetwork"], use_gpu=False)
        model = distributed_model_wrapper(model)
        opt = select_optimizer(
            model, config["NeuralNetwork"]["Training"]["Optimizer"])
        for _ in range(3):
            err, _ = train_fn(loaders[0], model, opt, 0)
        # all ranks must have identical params after DDP training
        p = torch.cat([x.flatten() for x in model.parameters()])
        plist = [torch.zeros_like(p) for _ in range(world_size)]
        dist.all_gather(plist, p)
        same = all(torch.allclose(plist[0], pi) for pi in plist)
        q.put((rank, bool(same), float(err)))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))


def test_ddp_gloo_training():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    world_size = 2
    port = 29531
    procs = [ctx.Process(target=_ddp_training_worker,
                         args=(r, world_size, port, q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(world_size)]
    for p in procs:
        p.join(timeout=60)
    for rank, same, err in results:
        assert same, f"rank {rank} param mismatch / error: {err}"


def _collectives_worker(rank, world_size, port, q):
    try:
        _init(rank, world_size, port)
        from hydragnn_amd.train import (gather_tensor_ranks,
                                        reduce_values_ranks)
        t = torch.full((1,), float(rank + 1))
        r = reduce_values_ranks(t.clone())
        # ranks have different row counts -> padded gather
        x = torch.arange((rank + 1) * 2, dtype=torch.float).view(-1, 1)
        g = gather_tensor_ranks(x)
        q.put((rank, float(r), g.shape[0], float(g.sum())))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, repr(e), None, None))


def test_metric_collectives():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_collectives_worker, args=(r, 2, 29532, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, r, n, s in results:
        assert isinstance(r, float), f"rank {rank} failed: {r}"
        assert r == pytest.approx(1.5)        # mean of 1 and 2
        assert n == 6                         # 2 + 4 rows gathered
        assert s == pytest.approx(0 + 1 + 0 + 1 + 2 + 3)


def _fsdp2_force_worker(rank, world_size, port, q):
    """FSDP2 + double-backward force path regression (reference
    tests/test_fsdp2_force_grad_regression.py)."""
    try:
        _init(rank, world_size, port)
        os.environ["HYDRAGNN_USE_FSDP"] = "1"
        os.environ["HYDRAGNN_FSDP_VERSION"] = "2"
        import sys
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        from hydragnn_amd.data import Batch
        from hydragnn_amd.models import create_model_config
        from hydragnn_amd.preprocess import create_dataloaders
        from hydragnn_amd.utils.config import update_config
        from hydragnn_amd.utils.datasets.synthetic import lj_dataset
        from hydragnn_amd.utils.distributed import get_distributed_model
        from deterministic_graph_data import base_config
        torch.manual_seed(1)
        config = base_config("SchNet", heads=("node",), num_epoch=1,
                             hidden_dim=16)
        arch = config["NeuralNetwork"]["Architecture"]
        arch.update({"enable_interatomic_potential": True,
                     "energy_weight": 1.0, "force_weight": 1.0,
                     "radius": 2.5})
        config["NeuralNetwork"]["Variables_of_interest"]["output_dim"] = [1]
        dataset = lj_dataset(num_samples=8, num_atoms=27, pbc=False)
        loaders = create_dataloaders(dataset, dataset, dataset, 4,
                                     config=config)
        config = update_config(config, *loaders)
        model = create_model_config(config["NeuralNetwork"], use_gpu=False)
        model = get_distributed_model(model)
        from hydragnn_amd.utils.distributed import set_reshard_after_backward
        set_reshard_after_backward(model, False)
        batch = Batch.from_data_list(dataset[:4])
        batch.pos.requires_grad_(True)
        pred = model(batch)
        loss, _ = model.module.energy_force_loss(pred, batch,
                                                 create_graph=True)
        loss.backward()
        ok = all(torch.isfinite(p.grad.to_local()
                                if hasattr(p.grad, "to_local")
                                else p.grad).all()
                 for p in model.parameters() if p.grad is not None)
        q.put((rank, bool(ok), float(loss)))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        os.environ.pop("HYDRAGNN_USE_FSDP", None)


def test_fsdp2_force_grad_regression():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_fsdp2_force_worker, args=(r, 2, 29533, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok, info in results:
        assert ok, f"rank {rank}: {info}"


def test_bench_two_rank_gloo():
    """The exact driver launch path: torchrun 2 ranks, CPU/gloo."""
    import subprocess
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    import sys
    cmd = [sys.executable,
         "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29541", os.path.join(repo, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--batch", "4"]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                       cwd=repo)
    assert r.returncode == 0, r.stderr[-3000:]
    import json
    line = [ln for ln in r.stdout.splitlines()
            if ln.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2
    assert d["config"]["global_batch"] == 8


def _mace_force_ddp_worker(rank, world_size, port, q):
    try:
        _init(rank, world_size, port)
        import sys
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        torch.manual_seed(11)
        from hydragnn_amd.data import Batch
        from hydragnn_amd.utils.datasets.synthetic import md17_shape_dataset
        from test_mace_model import _build, _mace_config
        ds = md17_shape_dataset(num_samples=8)
        model, config, _ = _build(_mace_config(), ds)
        ddp = torch.nn.parallel.DistributedDataParallel(model)
        opt = torch.optim.AdamW(ddp.parameters(), lr=1e-3)
        shard = ds[rank::world_size]
        for _ in range(2):
            batch = Batch.from_data_list([d.clone() for d in shard])
            batch.pos.requires_grad_(True)
            opt.zero_grad()
            pred = ddp(batch)
            loss, _ = model.energy_force_loss(pred, batch,
                                              create_graph=True)
            loss.backward()
            opt.step()
        p = torch.cat([x.flatten() for x in ddp.parameters()])
        plist = [torch.zeros_like(p) for _ in range(world_size)]
        dist.all_gather(plist, p)
        same = all(torch.allclose(plist[0], pi, atol=1e-6)
                   for pi in plist)
        q.put((rank, bool(same), float(loss)))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))


@pytest.mark.parametrize("world_size,port", [(2, 29531), (4, 29551),
                                             (8, 29561)])
def test_mace_force_training_ddp_gloo(world_size, port):
    """MACE att-interaction force training (double backward) under DDP
    on 2/4/8 gloo ranks (8 = the driver's full-node scaling shape):
    ranks end bitwise-synchronized."""
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_mace_force_ddp_worker,
                      args=(r, world_size, port, q))
          for r in range(world_size)]
    for p in ps:
        p.start()
    results = [q.get(timeout=600) for _ in range(world_size)]
    for p in ps:
        p.join(timeout=60)
    for rank, same, info in results:
        assert same, f"rank {rank}: {info}"


@pytest.mark.parametrize("world_size,port", [(4, 29571)])
def test_fsdp2_force_grad_regression_multirank(world_size, port):
    """FSDP2 double-backward force path on 4 gloo ranks."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_fsdp2_force_worker,
                         args=(r, world_size, port, q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = [q.get(timeout=600) for _ in range(world_size)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok, info in results:
        assert ok, f"rank {rank}: {info}"


def test_ddp_wrapper_find_unused_tristate():
    """None auto-enables find_unused_parameters for MLIP wrappers;
    an explicit False from the caller is respected."""
    import os

    import torch.distributed as dist

    from hydragnn_amd.utils.distributed import distributed_model_wrapper
    from test_mace_model import _build, _mace_config
    from hydragnn_amd.utils.datasets.synthetic import md17_shape_dataset

    import tempfile
    if dist.is_initialized():  # leftover group from another test
        dist.destroy_process_group()
    rdv = tempfile.NamedTemporaryFile(delete=False)
    dist.init_process_group("gloo", rank=0, world_size=1,
                            init_method=f"file://{rdv.name}")
    try:
        ds = md17_shape_dataset(num_samples=4)
        model, _, _ = _build(_mace_config(), ds)
        auto = distributed_model_wrapper(model)
        assert auto.find_unused_parameters is True
        model2, _, _ = _build(_mace_config(), ds)
        off = distributed_model_wrapper(model2,
                                        find_unused_parameters=False)
        assert off.find_unused_parameters is False
    finally:
        dist.destroy_process_group()
