"""Dataset layer: graph store roundtrip, pickle roundtrip, raw-format
parsers, cost-aware samplers (pattern: reference
tests/test_datasetclass_inheritance.py, test_cost_aware_batch_sampler)."""

import os

import numpy as np
import pytest
import torch

from hydragnn_amd.data import Batch, Data
from hydragnn_amd.preprocess import (
    CostAwareBatchSampler,
    DistributedCostAwareBatchSampler,
)
from hydragnn_amd.utils.datasets.graphstore import (
    DistDataset,
    GraphStoreDataset,
    GraphStoreWriter,
)
from hydragnn_amd.utils.datasets.pickledataset import (
    SimplePickleDataset,
    SimplePickleWriter,
)
from hydragnn_amd.utils.datasets.synthetic import lj_dataset


def _samples(n=6):
    return lj_dataset(num_samples=n, num_atoms=27, pbc=False)


def test_graphstore_roundtrip(tmp_path):
    ds = _samples()
    w = GraphStoreWriter("total", str(tmp_path))
    w.add(ds)
    w.add_global("pna_deg", [0, 1, 2, 3])
    w.add_global("minmax", np.array([[0.0], [1.0]]))
    w.save()

    r = GraphStoreDataset(str(tmp_path), "total")
    assert len(r) == len(ds)
    assert r.pna_deg == [0, 1, 2, 3]
    for i in (0, len(ds) - 1):
        a, b = ds[i], r[i]
        assert torch.allclose(a.pos, b.pos)
        assert torch.equal(a.edge_index, b.edge_index)
        assert torch.allclose(a.forces, b.forces)
    # preload + subset
    r2 = GraphStoreDataset(str(tmp_path), "total", preload=True,
                           subset=[1, 3])
    assert len(r2) == 2
    assert torch.allclose(r2[1].pos, ds[3].pos)
    assert r.get_node_counts() == [27] * len(ds)


def test_distdataset_shim(tmp_path):
    ds = _samples(4)
    w = GraphStoreWriter("total", str(tmp_path))
    w.add(ds)
    w.save()
    dd = DistDataset(str(tmp_path))
    dd.epoch_begin()
    s = dd[2]
    dd.epoch_end()
    assert torch.allclose(s.pos, ds[2].pos)
    assert dd.get_node_counts() == [27] * 4


def test_pickle_roundtrip(tmp_path):
    ds = _samples(4)
    SimplePickleWriter(ds, str(tmp_path), "total",
                       attrs={"pna_deg": [1, 2]})
    r = SimplePickleDataset(str(tmp_path), "total")
    assert len(r) == 4
    assert torch.allclose(r[1].pos, ds[1].pos)
    assert r.pna_deg == [1, 2]


def test_cost_aware_sampler():
    sizes = [5, 9, 3, 8, 2, 7, 6, 4]
    ds = []
    for s in sizes:
        d = Data(x=torch.zeros(s, 1))
        d.num_nodes = s
        ds.append(d)
    sampler = CostAwareBatchSampler(ds, max_nodes=10, shuffle=False)
    batches = list(sampler)
    for b in batches:
        assert sum(sizes[i] for i in b) <= 10
    assert sorted(i for b in batches for i in b) == list(range(len(sizes)))
    # deterministic under seed+epoch
    s2 = CostAwareBatchSampler(ds, max_nodes=10, shuffle=True, seed=1)
    s2.set_epoch(3)
    p1 = list(s2)
    s3 = CostAwareBatchSampler(ds, max_nodes=10, shuffle=True, seed=1)
    s3.set_epoch(3)
    assert p1 == list(s3)
    # oversized policies
    d_big = Data(x=torch.zeros(20, 1)); d_big.num_nodes = 20
    with pytest.raises(ValueError):
        list(CostAwareBatchSampler(ds + [d_big], max_nodes=10,
                                   shuffle=False))
    ok = CostAwareBatchSampler(ds + [d_big], max_nodes=10, shuffle=False,
                               oversized_policy="single")
    assert [len(ds)] in list(ok)


def test_distributed_cost_aware_identical_plan():
    sizes = [5, 9, 3, 8, 2, 7, 6, 4, 10, 1]
    ds = []
    for s in sizes:
        d = Data(x=torch.zeros(s, 1))
        d.num_nodes = s
        ds.append(d)
    s0 = DistributedCostAwareBatchSampler(ds, max_nodes=12, shuffle=True,
                                          seed=2, num_replicas=2, rank=0)
    s1 = DistributedCostAwareBatchSampler(ds, max_nodes=12, shuffle=True,
                                          seed=2, num_replicas=2, rank=1)
    s0.set_epoch(1); s1.set_epoch(1)
    b0, b1 = list(s0), list(s1)
    assert len(b0) == len(b1)  # equal steps on all ranks


def test_lsms_parser(tmp_path):
    raw = tmp_path / "raw"
    raw.mkdir()
    # graph features: 1 value at col 0; nodes: Z at col 5, charge col 6
    (raw / "sample1.txt").write_text(
        "1.5 0.0\n"
        "0 0 0.0 0.0 0.0 26 8.1\n"
        "1 0 1.0 0.0 0.0 26 7.9\n"
        "2 0 0.0 1.0 0.0 28 9.9\n")
    config = {
        "Dataset": {
            "name": "t", "format": "LSMS",
            "path": {"total": str(raw)},
            "node_features": {"name": ["Z", "charge"], "dim": [1, 1],
                              "column_index": [5, 6]},
            "graph_features": {"name": ["e"], "dim": [1],
                               "column_index": [0]},
        },
        "NeuralNetwork": {"Architecture": {
            "radius": 2.0, "max_neighbours": 10,
            "periodic_boundary_conditions": False}},
    }
    from hydragnn_amd.utils.datasets.rawloaders import LSMSDataset
    ds = LSMSDataset(config)
    assert len(ds) == 1
    d = ds[0]
    assert d.num_nodes == 3
    assert d.num_edges > 0
    assert d.get("edge_attr") is not None  # normalized edge lengths


def test_extxyz_parser(tmp_path):
    raw = tmp_path / "raw"
    raw.mkdir()
    (raw / "mol.xyz").write_text(
        '3\nLattice="5 0 0 0 5 0 0 0 5" energy=-7.2\n'
        "O 0.0 0.0 0.0\nH 0.96 0.0 0.0\nH 0.0 0.96 0.0\n")
    config = {
        "Dataset": {
            "name": "t", "format": "XYZ",
            "path": {"total": str(raw)},
            "node_features": {"name": ["Z"], "dim": [1],
                              "column_index": [0]},
            "graph_features": {"name": ["energy"], "dim": [1],
                               "column_index": [0]},
        },
        "NeuralNetwork": {"Architecture": {
            "radius": 1.5, "max_neighbours": 10,
            "periodic_boundary_conditions": False}},
    }
    from hydragnn_amd.utils.datasets.rawloaders import XYZDataset
    ds = XYZDataset(config)
    assert len(ds) == 1
    assert ds[0].num_nodes == 3
    assert ds[0].z.tolist() == [8, 1, 1]


def test_cfg_parser(tmp_path):
    raw = tmp_path / "raw"
    raw.mkdir()
    (raw / "c.cfg").write_text(
        "BEGIN_CFG\nSize\n2\nSupercell\n4 0 0\n0 4 0\n0 0 4\n"
        "AtomData: id type cartes_x cartes_y cartes_z fx fy fz\n"
        "1 13 0.0 0.0 0.0 0.1 0.0 0.0\n"
        "2 13 1.5 0.0 0.0 -0.1 0.0 0.0\n"
        "Energy\n-3.4\nEND_CFG\n")
    config = {
        "Dataset": {
            "name": "t", "format": "CFG",
            "path": {"total": str(raw)},
            "node_features": {"name": ["Z"], "dim": [1],
                              "column_index": [1]},
            "graph_features": {"name": ["energy"], "dim": [1],
                               "column_index": [0]},
        },
        "NeuralNetwork": {"Architecture": {
            "radius": 2.0, "max_neighbours": 10,
            "periodic_boundary_conditions": False}},
    }
    from hydragnn_amd.utils.datasets.rawloaders import CFGDataset
    ds = CFGDataset(config)
    assert len(ds) == 1
    assert ds[0].num_nodes == 2
    assert ds[0].forces.shape == (2, 3)


def test_graphstore_two_rank_write(tmp_path):
    """2-rank gloo write -> single reader sees both shards."""
    import subprocess, sys, textwrap
    script = tmp_path / "writer.py"
    script.write_text(textwrap.dedent(f"""
        import os, sys, torch, torch.distributed as dist
        sys.path.insert(0, {str(os.getcwd())!r})
        dist.init_process_group("gloo")
        rank = dist.get_rank()
        from hydragnn_amd.utils.datasets.graphstore import GraphStoreWriter
        from hydragnn_amd.utils.datasets.synthetic import lj_dataset
        ds = lj_dataset(num_samples=3, num_atoms=27, pbc=False,
                        seed=100 + rank)
        w = GraphStoreWriter("total", {str(tmp_path)!r})
        w.add(ds)
        w.save()
        dist.destroy_process_group()
    """))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29551", str(script)],
        capture_output=True, text=True, timeout=300, cwd=os.getcwd())
    assert r.returncode == 0, r.stderr[-1500:]
    reader = GraphStoreDataset(str(tmp_path), "total")
    assert len(reader) == 6
    # shard 0 and shard 1 differ (different seeds)
    assert not torch.allclose(reader[0].pos, reader[3].pos)


def test_scatter_max_cpu_argmax_backward():
    """CPU scatter-max backward routes gradient to the argmax edge."""
    src = torch.tensor([[1.0], [5.0], [3.0], [2.0]], requires_grad=True)
    idx = torch.tensor([0, 0, 1, 1])
    from hydragnn_amd.ops import scatter
    out = scatter(src, idx, 2, "max")
    assert out.flatten().tolist() == [5.0, 3.0]
    out.sum().backward()
    assert src.grad.flatten().tolist() == [0.0, 1.0, 1.0, 0.0]


def test_train_from_serialized_pkl_flow(tmp_path, monkeypatch):
    """End-to-end: total serialized pickle -> per-split pkls ->
    load_train_val_test_sets -> a short training run (reference
    test_datasetclass_inheritance pattern)."""
    import os
    import sys
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from deterministic_graph_data import (base_config,
                                          make_deterministic_dataset)
    from hydragnn_amd.models import create_model_config
    from hydragnn_amd.preprocess import (create_dataloaders,
                                         load_train_val_test_sets,
                                         total_to_train_val_test_pkls)
    from hydragnn_amd.train import train
    from hydragnn_amd.utils.config import update_config
    from hydragnn_amd.utils.datasets.serializeddataset import (
        SerializedWriter)
    from hydragnn_amd.utils.optimizer import select_optimizer

    monkeypatch.setenv("SERIALIZED_DATA_PATH", str(tmp_path))
    ds = make_deterministic_dataset(num_samples=40, num_heads_node=0)
    SerializedWriter(ds, f"{tmp_path}/serialized_dataset", "unit_test")
    config = base_config("GIN", heads=("graph",), num_epoch=2)
    config["Dataset"] = {"name": "unit_test", "path": {"total": "x"},
                         "format": "pickle"}
    config["NeuralNetwork"]["Training"]["perc_train"] = 0.7
    total_to_train_val_test_pkls(config)
    tr, va, te = load_train_val_test_sets(config)
    assert len(tr) == 28
    loaders = create_dataloaders(list(tr), list(va), list(te), 8,
                                 config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"], use_gpu=False)
    opt = select_optimizer(
        model, config["NeuralNetwork"]["Training"]["Optimizer"])
    first = None
    for _ in range(3):
        err, _ = train(loaders[0], model, opt, 0)
        first = first if first is not None else float(err)
    assert float(err) < first


def test_graph_dataset_prepare_pipeline(tmp_path):
    """load_pickled_graphs + prepare_graph_dataset: the historical
    three-object pickle -> edges rebuilt per config -> y/y_loc
    assembled (reference test_graph_dataset pattern)."""
    import pickle

    from hydragnn_amd.data import Data
    from hydragnn_amd.preprocess.graph_dataset import (
        load_and_prepare_graph_dataset)
    torch.manual_seed(0)
    samples = []
    for _ in range(6):
        pos = torch.rand(10, 3)
        x = torch.cat([torch.randint(1, 4, (10, 1)).float(),
                       torch.rand(10, 2)], dim=1)
        samples.append(Data(x=x, pos=pos,
                            y=torch.rand(3)))
    p = tmp_path / "total.pkl"
    with open(p, "wb") as f:
        pickle.dump(None, f)
        pickle.dump(None, f)
        pickle.dump(samples, f)
    config = {
        "NeuralNetwork": {
            "Architecture": {"radius": 0.8, "max_neighbours": 8,
                             "periodic_boundary_conditions": False},
            "Variables_of_interest": {
                "type": ["graph", "node"],
                "output_index": [0, 1],
                "input_node_features": [0],
            },
        },
        "Dataset": {
            "graph_features": {"dim": [1, 2]},
            "node_features": {"name": ["Z", "a", "b"],
                              "dim": [1, 1, 1],
                              "column_index": [0, 1, 2]},
        },
        "Verbosity": {"level": 0},
    }
    ds = load_and_prepare_graph_dataset(str(p), config)
    d = ds[0]
    assert d.edge_index.shape[0] == 2 and d.edge_index.shape[1] > 0
    # y = 1 graph value + 10 node values; y_loc = [0, 1, 11]
    assert d.y.shape[0] == 11
    assert d.y_loc.tolist() == [[0, 1, 11]]
    # input features restricted to column 0
    assert d.x.shape[1] == 1


def _sharded_fetch_worker(rank, world_size, port, q):
    try:
        import os

        import torch.distributed as dist
        os.environ.update({"MASTER_ADDR": "127.0.0.1",
                           "MASTER_PORT": str(port)})
        dist.init_process_group("gloo", rank=rank,
                                world_size=world_size)
        from hydragnn_amd.utils.datasets.graphstore import (
            ShardedDistDataset)
        from hydragnn_amd.utils.datasets.synthetic import lj_dataset
        # each rank owns a DIFFERENT shard (seed differs)
        local = lj_dataset(num_samples=3, num_atoms=6, pbc=False,
                           seed=100 + rank)
        ds = ShardedDistDataset(local)
        assert len(ds) == 3 * world_size
        ds.epoch_begin()
        # every rank reads EVERY global sample (local + remote)
        sums = []
        for i in range(len(ds)):
            d = ds.get(i)
            sums.append(float(d.pos.sum()))
        ds.epoch_end()
        # remote fetch outside the window must fail
        err = False
        try:
            ds.get((ds.offsets[rank] + 3) % len(ds))
        except RuntimeError:
            err = True
        # all ranks must agree on every sample
        t = torch.tensor(sums)
        ts = [torch.zeros_like(t) for _ in range(world_size)]
        dist.all_gather(ts, t)
        same = all(torch.allclose(ts[0], x) for x in ts)
        q.put((rank, bool(same and err), sums[:2]))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))


def test_sharded_dist_dataset_cross_rank_fetch():
    """DDStore-style data plane: remote samples fetched from their
    owner inside the epoch window, identical across ranks."""
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29537
    ps = [ctx.Process(target=_sharded_fetch_worker,
                      args=(r, 2, port, q)) for r in range(2)]
    for p in ps:
        p.start()
    results = [q.get(timeout=180) for _ in range(2)]
    for p in ps:
        p.join(timeout=60)
    for rank, ok, info in results:
        assert ok, f"rank {rank}: {info}"


def _sharded_train_worker(rank, world_size, port, q):
    try:
        import os
        import sys

        import torch.distributed as dist
        os.environ.update({"MASTER_ADDR": "127.0.0.1",
                           "MASTER_PORT": str(port)})
        dist.init_process_group("gloo", rank=rank,
                                world_size=world_size)
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        from deterministic_graph_data import (base_config,
                                              make_deterministic_dataset)
        from hydragnn_amd.models import create_model_config
        from hydragnn_amd.preprocess import create_dataloaders
        from hydragnn_amd.train import train as train_fn
        from hydragnn_amd.utils.config import update_config
        from hydragnn_amd.utils.datasets.graphstore import (
            ShardedDistDataset)
        from hydragnn_amd.utils.optimizer import select_optimizer
        torch.manual_seed(5)
        # each rank contributes a different shard to the global store
        full = make_deterministic_dataset(num_samples=16,
                                          num_heads_node=0)
        ds = ShardedDistDataset(full[rank::world_size])
        config = base_config("GIN", heads=("graph",), num_epoch=1)
        # config derivation iterates samples -> needs the fetch window
        ds.epoch_begin()
        loaders = create_dataloaders(ds, ds, ds, 4, config=config)
        config = update_config(config, *loaders)
        ds.epoch_end()
        model = create_model_config(config["NeuralNetwork"],
                                    use_gpu=False)
        opt = select_optimizer(
            model, config["NeuralNetwork"]["Training"]["Optimizer"])
        # the train loop opens/closes the fetch window itself
        err, _ = train_fn(loaders[0], model, opt, 0)
        q.put((rank, bool(err == err), float(err)))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))


def test_sharded_dataset_inside_train_loop():
    """The train loop drives the sharded store's epoch windows and
    every rank trains over the GLOBAL dataset (cross-rank fetches)."""
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29541
    ps = [ctx.Process(target=_sharded_train_worker,
                      args=(r, 2, port, q)) for r in range(2)]
    for p in ps:
        p.start()
    results = [q.get(timeout=240) for _ in range(2)]
    for p in ps:
        p.join(timeout=60)
    for rank, ok, info in results:
        assert ok, f"rank {rank}: {info}"


def test_compositional_stratified_splitting_behavior():
    """Every composition category lands in every split where possible,
    singletons are duplicated rather than dropped, and no sample is
    lost (reference compositional_data_splitting.py:19-156)."""
    import torch

    from hydragnn_amd.data import Data
    from hydragnn_amd.preprocess.compositional_splitting import (
        compositional_stratified_split, compositional_stratified_splitting,
        create_dataset_categories)

    def mol(zs):
        z = torch.tensor(zs, dtype=torch.long)
        d = Data(x=z.float().view(-1, 1), z=z,
                 pos=torch.randn(len(zs), 3),
                 y=torch.zeros(1, 1))
        d.num_nodes = len(zs)
        return d

    torch.manual_seed(0)
    # two well-populated compositions + one singleton
    ds = [mol([1, 1, 8]) for _ in range(10)] \
        + [mol([6, 6, 6, 1]) for _ in range(10)] + [mol([7, 7])]

    cats = create_dataset_categories(ds)
    assert len(set(cats)) == 3

    tr, va, te = compositional_stratified_splitting(ds, 0.8)
    # singleton duplicated -> 22 total, nothing lost
    assert len(tr) + len(va) + len(te) == len(ds) + 1

    def comps(split):
        return {tuple(sorted(set(d.z.tolist()))) for d in split}

    # every category reaches the train split
    assert comps(tr) == {(1, 8), (1, 6), (7,)}
    # the two populated categories appear in val+test too
    assert {(1, 8), (1, 6)} <= comps(va) | comps(te)

    # sklearn-backed entry point: proportional sizes, no loss
    tr2, va2, te2 = compositional_stratified_split(ds, 0.8, seed=1)
    assert len(tr2) + len(va2) + len(te2) == len(ds)
    assert abs(len(tr2) - 0.8 * len(ds)) <= 2


def test_md17_shape_targets_bounded():
    """Soft-cored LJ targets stay bounded even when the random
    geometry contains near-coincident pairs (unclamped LJ produced
    ~1e15 force targets and a ~1e30 initial loss that tripped the
    captured-step sanity check)."""
    import torch

    from hydragnn_amd.utils.datasets.synthetic import (
        md17_shape_dataset, md17_shape_dataset_fast)

    for ds in (md17_shape_dataset_fast(512, seed=13),
               md17_shape_dataset(64, seed=13)):
        y = torch.cat([d.y.reshape(-1) for d in ds])
        f = torch.cat([d.forces.reshape(-1) for d in ds])
        assert y.abs().max() < 1e3, float(y.abs().max())
        assert f.abs().max() < 1e5, float(f.abs().max())
        assert torch.isfinite(y).all() and torch.isfinite(f).all()
