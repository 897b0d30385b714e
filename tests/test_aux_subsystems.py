"""Aux subsystem unit tests: energy regression, download safety, LSMS
utils, descriptors, materials preprocessing, HPO helpers, custom
dataloader, tracer."""

import os

import numpy as np
import pytest
import torch

from hydragnn_amd.data import Data
from hydragnn_amd.preprocess import HydraDataLoader, parse_omp_places
from hydragnn_amd.preprocess.energy_linear_regression import (
    energy_linear_regression,
    shift_energies,
)
from hydragnn_amd.utils.datasets.download import safe_extract_tar, sha256_of
from hydragnn_amd.utils.datasets.synthetic import lj_dataset
from hydragnn_amd.utils.descriptors_and_embeddings import atomicdescriptors
from hydragnn_amd.utils.hpo.deephyper import (
    parse_slurm_nodelist,
    run_random_search,
)
from hydragnn_amd.utils.lsms import (
    compositional_histogram_cutoff,
    convert_raw_data_energy_to_gibbs,
)
from hydragnn_amd.utils.materials.preprocessing import (
    normalize_stress,
    validate_atomistic_sample,
    voigt_to_full,
)


def test_energy_linear_regression_recovers_reference_energies():
    g = torch.Generator().manual_seed(0)
    e_ref = {1: -13.6, 6: -1030.0, 8: -2040.0}
    ds = []
    for _ in range(30):
        z = torch.cat([
            torch.full((int(torch.randint(1, 5, (1,), generator=g)),), zz)
            for zz in (1, 6, 8)]).long()
        e = sum(e_ref[int(v)] for v in z) + float(
            torch.randn(1, generator=g)) * 1e-3
        d = Data(z=z, x=z.float().view(-1, 1),
                 energy=torch.tensor([[e]]), y=torch.tensor([[e]]))
        d.num_nodes = z.numel()
        ds.append(d)
    e_fit, present = energy_linear_regression(ds, distributed=False)
    for zz, val in e_ref.items():
        assert abs(e_fit[zz] - val) < 0.05, (zz, e_fit[zz])
    shift_energies(ds, e_fit)
    assert abs(float(ds[0].energy)) < 1.0


def test_safe_tar_rejects_traversal(tmp_path):
    import tarfile
    import io
    bad = tmp_path / "bad.tar"
    with tarfile.open(bad, "w") as tar:
        info = tarfile.TarInfo("../evil.txt")
        data = b"x"
        info.size = len(data)
        tar.addfile(info, io.BytesIO(data))
    with pytest.raises(ValueError):
        safe_extract_tar(str(bad), str(tmp_path / "out"))


def test_sha256(tmp_path):
    p = tmp_path / "f.bin"
    p.write_bytes(b"hello")
    assert sha256_of(str(p)).startswith("2cf24dba")


def test_lsms_utils():
    ds = lj_dataset(num_samples=6, num_atoms=27, pbc=False)
    pure = {13: float(ds[0].y) / 27}
    convert_raw_data_energy_to_gibbs(ds, pure)
    assert abs(float(ds[0].y)) < 1e-4  # formation energy of "pure" = 0
    kept = compositional_histogram_cutoff(ds, 13, max_per_bin=2)
    assert len(kept) == 2


def test_atomic_descriptors():
    ad = atomicdescriptors(element_types=[1, 6, 8])
    f = ad.get_atom_features(torch.tensor([1, 6, 8]))
    assert f.shape == (3, 3 + 5)
    assert f[0, 0] == 1.0 and f[1, 1] == 1.0


def test_materials_preprocessing():
    s = voigt_to_full(torch.tensor([1., 2., 3., 4., 5., 6.]))
    assert s[0, 0] == 1 and s[1, 2] == 4 and s[0, 1] == 6
    n = normalize_stress(torch.eye(3), units="GPa")
    assert abs(float(n[0, 0]) - 1 / 160.21766208) < 1e-9
    d = lj_dataset(num_samples=1, num_atoms=27, pbc=False)[0]
    validate_atomistic_sample(d, require_forces=True)
    bad = d.clone()
    bad.forces = bad.forces[:3]
    with pytest.raises(ValueError, match="forces"):
        validate_atomistic_sample(bad, require_forces=True)


def test_hpo_helpers():
    hosts = parse_slurm_nodelist("node[001-003,007]")
    assert hosts == ["node001", "node002", "node003", "node007"]
    best, val, hist = run_random_search(
        lambda c: (c["lr"] - 0.01) ** 2,
        {"lr": (1e-4, 1e-1, "log"), "dim": [16, 32]}, num_trials=20)
    assert len(hist) == 20 and val < 0.01


def test_parse_omp_places():
    assert parse_omp_places("{0},{1},{2}") == [0, 1, 2]
    assert parse_omp_places("{0:4}") == [0, 1, 2, 3]


def test_hydra_dataloader_matches_default():
    ds = lj_dataset(num_samples=10, num_atoms=8, pbc=False)
    loader = HydraDataLoader(ds, batch_size=4, shuffle=False,
                             num_workers=2)
    batches = list(loader)
    assert len(loader) == 3 and len(batches) == 3
    assert batches[0].num_graphs == 4
    assert batches[-1].num_graphs == 2
    assert torch.allclose(batches[0].pos[:8], ds[0].pos)


def test_tracer_roundtrip(tmp_path):
    from hydragnn_amd.utils.profiling_and_tracing import tracer as tr
    tr.reset()
    tr.enable()
    tr.start("region")
    tr.stop("region")
    tr.save(str(tmp_path))
    assert (tmp_path / "gp_timing.p0").exists()
    tr.disable()
    tr.reset()


def test_download_with_local_http_server(tmp_path):
    """Resumable sha256-verified download against a local HTTP server
    (reference tests/test_dataset_download* pattern)."""
    import http.server
    import threading
    from functools import partial
    from hydragnn_amd.utils.datasets.download import download, sha256_of

    src = tmp_path / "serve"
    src.mkdir()
    payload = bytes(range(256)) * 4096  # 1 MiB
    (src / "data.bin").write_bytes(payload)
    handler = partial(http.server.SimpleHTTPRequestHandler,
                      directory=str(src))
    httpd = http.server.ThreadingHTTPServer(("127.0.0.1", 0), handler)
    port = httpd.server_address[1]
    t = threading.Thread(target=httpd.serve_forever, daemon=True)
    t.start()
    try:
        dest = tmp_path / "out" / "data.bin"
        url = f"http://127.0.0.1:{port}/data.bin"
        sha = sha256_of(str(src / "data.bin"))
        download(url, str(dest), sha)
        assert dest.read_bytes() == payload
        # resume: truncate to a .part and re-download (Range request)
        dest.unlink()
        part = tmp_path / "out" / "data.bin.part"
        part.write_bytes(payload[: len(payload) // 2])
        download(url, str(dest), sha)
        assert dest.read_bytes() == payload
        # wrong checksum rejected
        dest.unlink()
        with pytest.raises(ValueError, match="sha256"):
            download(url, str(dest), "0" * 64)
    finally:
        httpd.shutdown()


def test_radial_bases():
    """Radial basis properties (reference test_radial_transforms
    pattern): cutoff zeros, limits, shapes."""
    from hydragnn_amd.ops import (bessel_basis, chebyshev_basis,
                                  cosine_cutoff, gaussian_basis,
                                  polynomial_cutoff, sinc_basis)
    import math
    r = torch.linspace(0.01, 6.0, 50).view(-1, 1)
    w = torch.arange(1, 9).float() * math.pi / 5.0
    b = bessel_basis(r, 5.0, w)
    assert b.shape == (50, 8)
    # bessel bases vanish at r_max (sin(n*pi) = 0)
    at_rmax = bessel_basis(torch.tensor([[5.0]]), 5.0, w)
    assert at_rmax.abs().max() < 1e-5
    # polynomial cutoff: 1 at 0, 0 beyond r_max, monotone-ish
    pc = polynomial_cutoff(r, 5.0)
    assert float(polynomial_cutoff(torch.tensor([[0.0]]), 5.0)) == pytest.approx(1.0)
    assert float(polynomial_cutoff(torch.tensor([[5.5]]), 5.0)) == 0.0
    cc = cosine_cutoff(r.squeeze(-1), 5.0)
    assert float(cosine_cutoff(torch.tensor([0.0]), 5.0)) == pytest.approx(1.0)
    assert float(cosine_cutoff(torch.tensor([5.1]), 5.0)) == 0.0
    # sinc basis finite at r -> 0 with the analytic limit n*pi/rc
    s0 = sinc_basis(torch.tensor([[1e-12]]), 5.0, 4)
    assert torch.allclose(
        s0.flatten(),
        torch.arange(1, 5).float() * math.pi / 5.0, atol=1e-4)
    ch = chebyshev_basis(r, 6.0, 5)
    assert ch.shape == (50, 5)
    assert (ch.abs() <= 1.0 + 1e-6).all()
    g = gaussian_basis(r, torch.linspace(0, 5, 10).view(1, -1), -0.5)
    assert g.shape == (50, 10) and (g <= 1.0).all()


def test_formation_enthalpy_two_components():
    """Reference test_enthalpy pattern: linear-mixing energies of a
    2-component system give ~zero formation enthalpy against the pure
    references."""
    from hydragnn_amd.utils.lsms import get_formation_enthalpy
    e_pure = {13: -3.5, 29: -4.2}
    # perfectly linear mixture: E_total = sum n_i * e_pure_i
    for n_al, n_cu in [(4, 0), (0, 4), (3, 1), (2, 2)]:
        comp = {13: n_al, 29: n_cu}
        e_tot = n_al * e_pure[13] + n_cu * e_pure[29]
        h = get_formation_enthalpy(e_tot, comp, e_pure)
        assert abs(h) < 1e-10
    # non-linear mixing shows up as the total excess energy
    h = get_formation_enthalpy(2 * e_pure[13] + 2 * e_pure[29] - 0.4,
                               {13: 2, 29: 2}, e_pure)
    assert h == pytest.approx(-0.4)


def test_parse_deepspeed_config():
    from hydragnn_amd.utils.config import parse_deepspeed_config
    cfg = {"NeuralNetwork": {"Training": {
        "batch_size": 16,
        "Optimizer": {"type": "AdamW", "learning_rate": 5e-4}}}}
    ds = parse_deepspeed_config(cfg)
    # per-GPU micro batch (a global train_batch_size equal to the
    # local batch fails DeepSpeed's consistency check at world > 1)
    assert ds["train_micro_batch_size_per_gpu"] == 16
    assert ds["gradient_accumulation_steps"] == 1
    assert ds["steps_per_print"] >= 1e9
    # no optimizer section: the wrapper passes the built optimizer
    # instance to deepspeed.initialize, and both together are rejected
    assert "optimizer" not in ds
    # user ds_config passes through untouched
    cfg["NeuralNetwork"]["ds_config"] = {
        "train_micro_batch_size_per_gpu": 4, "zero_optimization":
        {"stage": 1}}
    ds = parse_deepspeed_config(cfg)
    assert ds["train_micro_batch_size_per_gpu"] == 4
    assert ds["zero_optimization"]["stage"] == 1
    assert "gradient_accumulation_steps" not in ds


def test_visualizer_plot_suite(tmp_path):
    """All visualizer plot types render to files (reference
    visualizer coverage pattern)."""
    pytest.importorskip("matplotlib")
    from hydragnn_amd.postprocess.visualizer import Visualizer
    torch.manual_seed(0)
    viz = Visualizer("viz_test", num_heads=2, path=str(tmp_path))
    t = [torch.randn(50, 1), torch.randn(50, 1)]
    p = [x + 0.1 * torch.randn_like(x) for x in t]
    viz.add_history(1.0, 1.1, 1.2)
    viz.add_history(0.5, 0.6, 0.7)
    viz.plot_history()
    viz.create_scatter_plots(t, p)
    viz.create_error_histograms(t, p)
    viz.create_plot_global_analysis(t, p)
    nc = torch.randint(4, 8, (50,))
    viz.create_error_histogram_per_node(t, p, nc)
    viz.create_parity_plot_vector(torch.randn(30, 3),
                                  torch.randn(30, 3))
    viz.num_nodes_plot(nc)
    viz.create_plot_global(t, p, output_names=["a", "b"])
    viz.create_parity_plot_and_error_histogram_scalar(
        "energy", t[0], p[0], iepoch=1)
    viz.create_parity_plot_and_error_histogram_scalar(
        "pernode", torch.randn(20, 6), torch.randn(20, 6))
    viz.create_parity_plot_per_node_vector(
        "forces", torch.randn(10, 4, 3), torch.randn(10, 4, 3), 4)
    out = tmp_path / "viz_test"
    for f in ["history.png", "scatter.png", "error_hist.png",
              "global_analysis.png", "error_hist_per_size.png",
              "parity_forces.png", "num_nodes.png",
              "parity_global.png", "parity_hist_energy_epoch1.png",
              "parity_hist_pernode.png",
              "parity_pernode_forces.png"]:
        assert (out / f).exists(), f


def test_xyz_to_graph_bond_perception():
    """Own bond perception (the vendored-xyz2mol role): water, CO2 and
    N2 get chemically correct graphs and bond orders."""
    from hydragnn_amd.utils.descriptors_and_embeddings.xyz2graph import (
        assign_bond_orders, perceive_bonds, xyz_to_graph)

    # water: two O-H single bonds, no H-H bond
    z = torch.tensor([8, 1, 1])
    pos = torch.tensor([[0.0, 0.0, 0.0], [0.96, 0.0, 0.0],
                        [-0.24, 0.93, 0.0]])
    bonds, lengths = perceive_bonds(z, pos)
    assert sorted(map(tuple, bonds.tolist())) == [(0, 1), (0, 2)]
    orders, charges = assign_bond_orders(z, bonds, lengths)
    assert orders.tolist() == [1, 1]
    assert charges.abs().sum() == 0

    # CO2: two C=O double bonds
    z = torch.tensor([6, 8, 8])
    pos = torch.tensor([[0.0, 0.0, 0.0], [1.16, 0.0, 0.0],
                        [-1.16, 0.0, 0.0]])
    bonds, lengths = perceive_bonds(z, pos)
    orders, _ = assign_bond_orders(z, bonds, lengths)
    assert sorted(orders.tolist()) == [2, 2]

    # N2: triple bond
    z = torch.tensor([7, 7])
    pos = torch.tensor([[0.0, 0.0, 0.0], [1.10, 0.0, 0.0]])
    bonds, lengths = perceive_bonds(z, pos)
    orders, _ = assign_bond_orders(z, bonds, lengths)
    assert orders.tolist() == [3]

    # full pipeline -> trainable Data with dst-sorted edges
    d = xyz_to_graph(torch.tensor([6, 1, 1, 1, 1]), torch.tensor(
        [[0.0, 0, 0], [1.09, 0, 0], [-0.36, 1.03, 0],
         [-0.36, -0.51, 0.89], [-0.36, -0.51, -0.89]]))
    assert d.num_edges == 8  # 4 C-H bonds, both directions
    dst = d.edge_index[1]
    assert bool((dst[1:] >= dst[:-1]).all())
    assert d.edge_attr.shape == (8, 2)


def test_oversampling_dataloader():
    """oversampling=True draws num_samples per epoch with replacement
    (reference load_data.py RandomSampler path)."""
    import sys as _sys
    _sys.path.insert(0, os.path.dirname(__file__))
    from deterministic_graph_data import make_deterministic_dataset
    from hydragnn_amd.preprocess import create_dataloaders

    ds = make_deterministic_dataset(num_samples=6, num_heads_node=0)
    tr, va, te = create_dataloaders(ds, ds, ds, batch_size=4,
                                    oversampling=True, num_samples=20)
    seen = sum(b.num_graphs for b in tr)
    assert seen == 20


def _zero_worker(rank, world, port, q):
    try:
        os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                          RANK=str(rank), WORLD_SIZE=str(world),
                          LOCAL_RANK=str(rank))
        import torch.distributed as dist

        from hydragnn_amd.utils.optimizer import select_optimizer
        dist.init_process_group("gloo", rank=rank, world_size=world)
        torch.manual_seed(0)
        m = torch.nn.Linear(4, 4)
        opt = select_optimizer(m, {"type": "AdamW",
                                   "learning_rate": 1e-3,
                                   "use_zero_redundancy": True})
        from torch.distributed.optim import ZeroRedundancyOptimizer
        assert isinstance(opt, ZeroRedundancyOptimizer)
        torch.manual_seed(10 + rank)
        m(torch.randn(2, 4)).sum().backward()
        opt.step()
        opt.consolidate_state_dict(to=0)
        if rank == 0:
            sd = opt.state_dict()
            assert "state" in sd and len(sd["state"]) == 2
        dist.destroy_process_group()
        q.put((rank, True, ""))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))


def test_zero_redundancy_optimizer_two_rank():
    """use_zero_redundancy shards optimizer state at world > 1 and
    consolidates for checkpointing (reference optimizer.py:53-111)."""
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_zero_worker, args=(r, 2, 29731, q))
          for r in range(2)]
    for p in ps:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in ps:
        p.join(timeout=60)
    for rank, ok, info in results:
        assert ok, f"rank {rank}: {info}"
