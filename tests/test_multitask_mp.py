"""Branch/task model-parallel training over 2 gloo ranks (pattern:
reference examples/multibranch/train.py:230-546 driver)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

pytestmark = pytest.mark.mpi


def _worker(rank, world_size, port, q):
    try:
        os.environ.update({
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
            "RANK": str(rank), "WORLD_SIZE": str(world_size),
            "LOCAL_RANK": str(rank),
        })
        dist.init_process_group("gloo", rank=rank, world_size=world_size)
        import sys
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        torch.manual_seed(7)
        from deterministic_graph_data import (base_config,
                                              make_deterministic_dataset)
        from hydragnn_amd.data import Batch
        from hydragnn_amd.models import create_model_config
        from hydragnn_amd.models.multitask_mp import MultiTaskModelMP
        from hydragnn_amd.preprocess import create_dataloaders
        from hydragnn_amd.utils.config import update_config

        # 2 branches, each rank = one branch color
        nbranch = 2
        mycolor = rank % nbranch
        branch_group = None
        for color in range(nbranch):
            ranks = [r for r in range(world_size)
                     if r % nbranch == color]
            g = dist.new_group(ranks=ranks)
            if color == mycolor:
                branch_group = g

        config = base_config("GIN", heads=("graph",), num_epoch=1)
        arch = config["NeuralNetwork"]["Architecture"]
        # two graph branches
        arch["output_heads"] = {"graph": [
            {"type": "branch-0", "architecture": {
                "num_sharedlayers": 1, "dim_sharedlayers": 8,
                "num_headlayers": 1, "dim_headlayers": [8]}},
            {"type": "branch-1", "architecture": {
                "num_sharedlayers": 1, "dim_sharedlayers": 8,
                "num_headlayers": 1, "dim_headlayers": [8]}},
        ]}
        dataset = make_deterministic_dataset(num_samples=16,
                                             num_heads_node=0)
        for d in dataset:
            d.dataset_name = torch.tensor([[mycolor]])
        loaders = create_dataloaders(dataset, dataset, dataset, 8,
                                     config=config)
        config = update_config(config, *loaders)
        base = create_model_config(config["NeuralNetwork"], use_gpu=False)
        model = MultiTaskModelMP(base, mycolor, branch_group)
        opt = model.make_dual_optimizer(lr=0.01)

        losses = []
        for _ in range(4):
            tot = 0.0
            for batch in loaders[0]:
                opt.zero_grad()
                pred = model(batch)
                head_index = [torch.arange(batch.y.shape[0])]
                loss, _ = model.loss(pred, batch.y, head_index)
                loss.backward()
                opt.step()
                tot += float(loss)
            losses.append(tot)

        # encoder params identical across WORLD
        enc = model.encoder.module if hasattr(model.encoder, "module") \
            else model.encoder
        p = torch.cat([x.flatten() for x in enc.parameters()])
        plist = [torch.zeros_like(p) for _ in range(world_size)]
        dist.all_gather(plist, p)
        enc_same = all(torch.allclose(plist[0], pi) for pi in plist)
        q.put((rank, bool(enc_same), losses[-1] < losses[0]))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        import traceback
        q.put((rank, False, traceback.format_exc()))


def test_multitask_branch_parallel():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29534, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, enc_same, decreasing in results:
        assert enc_same is True, f"rank {rank}: {decreasing}"
        assert decreasing, f"rank {rank}: loss did not decrease"
