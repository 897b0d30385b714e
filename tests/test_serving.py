"""HTTP inference serving (hydragnn_amd/serve.py): health, graph-head
predictions, and MLIP energy+forces over FastAPI's test client."""

import os
import sys

import pytest
import torch

pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def _sample(n=6, seed=0):
    g = torch.Generator().manual_seed(seed)
    return {"pos": (torch.rand(n, 3, generator=g) * 2).tolist(),
            "z": [1] * n}


def test_serving_graph_head():
    from _training_workflow import run_training
    from hydragnn_amd.serve import create_app
    model, config, _ = run_training("GIN", heads=("graph",),
                                    num_samples=16, num_epoch=2)
    app = create_app(model, config, device="cpu")
    client = TestClient(app)
    r = client.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"
    r = client.post("/predict", json={"samples": [_sample(), _sample(8, 1)],
                                      "radius": 1.5})
    assert r.status_code == 200, r.text
    heads = r.json()["heads"]
    assert len(heads) == 1 and len(heads[0]) == 2  # one head, 2 graphs


def test_serving_mlip_energy_forces():
    from test_mace_model import _build, _mace_config
    from hydragnn_amd.serve import create_app
    from hydragnn_amd.utils.datasets.synthetic import md17_shape_dataset
    ds = md17_shape_dataset(num_samples=4)
    model, config, _ = _build(_mace_config(), ds)
    app = create_app(model, device="cpu")
    client = TestClient(app)
    assert client.get("/health").json()["mlip"]
    body = {"samples": [_sample(5, 2)], "radius": 3.0,
            "max_neighbours": 10}
    r = client.post("/predict", json=body)
    assert r.status_code == 200, r.text
    out = r.json()
    assert len(out["energy"]) == 1
    assert len(out["forces"][0]) == 5
    assert len(out["forces"][0][0]) == 3
    # malformed input -> 400
    r = client.post("/predict", json={"samples": [{"pos": "bad"}]})
    assert r.status_code == 400


@pytest.mark.gpu
def test_serving_mlip_on_gpu():
    """Serving runs MLIP energy+forces on cuda:0 (device autoselect)."""
    if not torch.cuda.is_available():
        pytest.skip("requires GPU")
    from test_mace_model import _build, _mace_config
    from hydragnn_amd.utils.datasets.synthetic import md17_shape_dataset
    from hydragnn_amd.serve import create_app
    ds = md17_shape_dataset(num_samples=8)
    model, config, _ = _build(_mace_config(), ds)
    app = create_app(model, config)  # autoselects cuda
    client = TestClient(app)
    r = client.get("/health")
    assert r.json()["device"].startswith("cuda")
    r = client.post("/predict", json={"samples": [_sample(6, 2)],
                                      "radius": 7.0})
    assert r.status_code == 200, r.text
    body = r.json()
    assert "energy" in body and "forces" in body
    assert all(e == e for e in body["energy"])
