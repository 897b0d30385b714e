"""Inference serving for trained models.

The task brief calls for production deployment AND serving; the
reference stops at run_prediction.  This module exposes a trained
model over HTTP (FastAPI + uvicorn, both in the image): clients POST
atomic structures, the server builds radius graphs with the same ops
the training path uses and returns per-head predictions — and, for
interatomic-potential models, energies and forces.

    python -m hydragnn_amd.serve --config examples/md17/md17_mlip.json \
        --checkpoint logs/<name>  [--host 0.0.0.0 --port 8000]

Request schema (POST /predict):
    {"samples": [{"pos": [[x,y,z],...], "z": [Z,...] | "x": [[...],...],
                  "cell": [[...]x3] optional, "pbc": [true,true,true]}],
     "radius": optional override, "max_neighbours": optional}
"""

import argparse
from typing import List, Optional

import torch

try:  # request schema at module scope so FastAPI can resolve the hint
    from pydantic import BaseModel

    class PredictRequest(BaseModel):
        samples: List[dict]
        radius: Optional[float] = None
        max_neighbours: Optional[int] = None
except ImportError:  # pragma: no cover - serving extras absent
    PredictRequest = None


def _build_sample(spec: dict, radius: float, max_neighbours: int):
    from .data import Data
    from .ops import radius_graph, radius_graph_pbc
    pos = torch.tensor(spec["pos"], dtype=torch.get_default_dtype())
    if "x" in spec:
        x = torch.tensor(spec["x"], dtype=torch.get_default_dtype())
        if x.dim() == 1:
            x = x.view(-1, 1)
    elif "z" in spec:
        x = torch.tensor(spec["z"],
                         dtype=torch.get_default_dtype()).view(-1, 1)
    else:
        x = torch.ones(pos.shape[0], 1)
    d = Data(x=x, pos=pos, y=torch.zeros(1, 1))
    if "z" in spec:
        d.z = torch.tensor(spec["z"], dtype=torch.long)
    if spec.get("cell") is not None:
        d.cell = torch.tensor(spec["cell"],
                              dtype=torch.get_default_dtype())
        d.pbc = tuple(spec.get("pbc", (True, True, True)))
        d.edge_index, d.edge_shifts = radius_graph_pbc(
            d.pos, radius, d.cell, pbc=d.pbc,
            max_num_neighbors=max_neighbours)
    else:
        d.edge_index = radius_graph(d.pos, radius,
                                    max_num_neighbors=max_neighbours)
    return d


def create_app(model, config: Optional[dict] = None,
               device: Optional[str] = None):
    """FastAPI app serving `model` (a trained hydragnn_amd model or
    MLIP wrapper)."""
    from fastapi import FastAPI, HTTPException

    from .data import Batch

    arch = ((config or {}).get("NeuralNetwork", {})
            .get("Architecture", {}))
    default_radius = float(arch.get("radius", 5.0))
    default_nbr = int(arch.get("max_neighbours", 32))
    dev = device or ("cuda" if torch.cuda.is_available() else "cpu")
    core = model.module if hasattr(model, "module") else model
    core = core.to(dev).eval()
    is_mlip = hasattr(core, "energy_force_loss")

    app = FastAPI(title="hydragnn_amd inference",
                  description="graph-property / MLIP serving")

    @app.get("/health")
    def health():
        return {"status": "ok", "device": dev, "mlip": is_mlip}

    @app.post("/predict")
    def predict(req: PredictRequest):
        if not req.samples:
            raise HTTPException(400, "no samples")
        r = req.radius or default_radius
        nbr = req.max_neighbours or default_nbr
        try:
            data_list = [_build_sample(s, r, nbr) for s in req.samples]
        except (KeyError, TypeError, ValueError) as e:
            raise HTTPException(400, f"bad sample: {e}")
        batch = Batch.from_data_list(data_list).to(dev)
        counts = [int(d.num_nodes) for d in data_list]
        if is_mlip:
            from .ops import scatter
            batch.pos.requires_grad_(True)
            pred = core(batch)
            # head 0 is per-atom energy: total energy = per-graph sum
            node_e = pred[0].reshape(-1)
            energy = scatter(node_e, batch.batch,
                             len(counts), "sum")
            forces = -torch.autograd.grad(
                energy.sum(), batch.pos)[0]
            out = {"energy": energy.detach().cpu().tolist(),
                   "forces": []}
            off = 0
            for n in counts:
                out["forces"].append(
                    forces[off:off + n].detach().cpu().tolist())
                off += n
            return out
        with torch.no_grad():
            pred = core(batch)
        heads = []
        for ihead in range(core.num_heads):
            p = pred[0][ihead] if core.var_output else pred[ihead]
            heads.append(p.detach().cpu().tolist())
        return {"heads": heads}

    return app


def main():
    import json

    import uvicorn

    from .models import create_model_config
    from .preprocess import create_dataloaders
    from .utils.config import update_config
    from .utils.datasets.synthetic import lj_dataset
    from .utils.model import load_existing_model

    import os

    parser = argparse.ArgumentParser()
    parser.add_argument("--config", default=None,
                        help="raw config JSON (only needed when the "
                             "checkpoint dir has no saved config.json)")
    parser.add_argument("--checkpoint", default=None,
                        help="log name to load the .pk checkpoint from")
    parser.add_argument("--logdir", default="./logs/")
    parser.add_argument("--host", default="127.0.0.1")
    parser.add_argument("--port", type=int, default=8000)
    args = parser.parse_args()

    # Prefer the config.json that training saved next to the checkpoint
    # (save_config): it already carries the dataset-derived fields
    # (output dims from y_loc, pna_deg, ...), so the architecture built
    # here matches the checkpointed weights exactly.
    saved_cfg = (os.path.join(args.logdir, args.checkpoint, "config.json")
                 if args.checkpoint else None)
    if saved_cfg and os.path.exists(saved_cfg):
        with open(saved_cfg) as f:
            config = json.load(f)
    elif args.config:
        with open(args.config) as f:
            config = json.load(f)
        if "output_dim" not in config["NeuralNetwork"]["Architecture"]:
            # raw config: derive output dims the way training does,
            # over a tiny synthetic set (shape-only; weights come from
            # the checkpoint)
            ds = lj_dataset(num_samples=8, num_atoms=8, pbc=False)
            loaders = create_dataloaders(
                ds, ds, ds,
                config["NeuralNetwork"]["Training"]["batch_size"],
                config=config)
            config = update_config(config, *loaders)
    else:
        raise SystemExit("need --config or a checkpoint dir with a "
                         "saved config.json")
    model = create_model_config(config["NeuralNetwork"])
    if args.checkpoint:
        load_existing_model(model, args.checkpoint, path=args.logdir)
    app = create_app(model, config)
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
