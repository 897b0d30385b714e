"""Cross-feature combination matrix: pairs of features that interact
through shared plumbing (attention engines x multihead, precision x
stacks, pooling variants, edge features) — each a forward (+backward
where relevant) on CPU."""

import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from deterministic_graph_data import base_config, make_deterministic_dataset  # noqa: E402
from hydragnn_amd.data import Batch  # noqa: E402
from hydragnn_amd.models import create_model_config  # noqa: E402
from hydragnn_amd.preprocess import add_laplacian_pe, create_dataloaders  # noqa: E402
from hydragnn_amd.utils.config import update_config  # noqa: E402


def _build(mpnn, arch_extra=None, heads=("graph",), node_heads=0,
           pe=False):
    cfg = base_config(mpnn, heads=heads, num_epoch=1)
    if arch_extra:
        cfg["NeuralNetwork"]["Architecture"].update(arch_extra)
    ds = make_deterministic_dataset(num_samples=8,
                                    num_heads_node=node_heads)
    if pe:
        for d in ds:
            add_laplacian_pe(d, 3)
    loaders = create_dataloaders(ds, ds, ds, 4, config=cfg)
    cfg = update_config(cfg, *loaders)
    m = create_model_config(cfg["NeuralNetwork"], use_gpu=False)
    return m, Batch.from_data_list(ds[:4])


def test_gps_with_graph_and_node_heads():
    m, b = _build("GIN", {"global_attn_engine": "gps",
                          "global_attn_heads": 4, "pe_dim": 3},
                  heads=("graph", "node"), node_heads=1, pe=True)
    out = m(b)
    assert len(out) == 2


def test_mace_fp64_forward_backward():
    prev = torch.get_default_dtype()
    torch.set_default_dtype(torch.float64)
    try:
        m, b = _build("MACE", {"max_ell": 2, "node_max_ell": 1,
                               "correlation": 2, "num_radial": 8})
        m = m.double()
        b.pos = b.pos.double()
        b.x = b.x.double()
        out = m(b)
        assert out[0].dtype == torch.float64
        out[0].sum().backward()
        assert all(torch.isfinite(p.grad).all()
                   for p in m.parameters() if p.grad is not None)
    finally:
        torch.set_default_dtype(prev)


@pytest.mark.parametrize("pooling", ["max", "add"])
def test_pooling_variants(pooling):
    m, b = _build("PNAEq", {"graph_pooling": pooling})
    assert torch.isfinite(m(b)[0]).all()


def test_mace_multibranch_decode():
    cfg = base_config("MACE", heads=("graph",), num_epoch=1)
    arch = cfg["NeuralNetwork"]["Architecture"]
    arch.update({"max_ell": 2, "node_max_ell": 1, "correlation": 2,
                 "num_radial": 8})
    arch["output_heads"]["graph"] = [
        {"type": f"branch-{bidx}", "architecture":
         {"num_sharedlayers": 1, "dim_sharedlayers": 8,
          "num_headlayers": 1, "dim_headlayers": [8]}}
        for bidx in range(2)]
    ds = make_deterministic_dataset(num_samples=8, num_heads_node=0)
    for i, d in enumerate(ds):
        d.dataset_name = torch.tensor([i % 2])
    loaders = create_dataloaders(ds, ds, ds, 4, config=cfg)
    cfg = update_config(cfg, *loaders)
    m = create_model_config(cfg["NeuralNetwork"], use_gpu=False)
    out = m(Batch.from_data_list(ds[:4]))
    assert out[0].shape == (4, 1)
