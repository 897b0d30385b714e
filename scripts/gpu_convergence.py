"""MACE MD17-shape force training to convergence on the MI355X
(VERDICT r1 item 4): bf16 training on gfx950 must demonstrably
converge, not just compute.  Writes the per-epoch loss / energy-MAE /
force-MAE trajectory to gpurun_out/convergence_mace_md17.json
(committed under profiles/ after the run)."""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch  # noqa: E402


def main():
    from torch.utils.data import DataLoader

    import bench as B
    from hydragnn_amd.preprocess.load_data import _collate
    from hydragnn_amd.train import train
    from hydragnn_amd.utils.datasets.synthetic import (
        md17_shape_dataset_fast)

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    precision = os.environ.get("CONV_PRECISION", "bf16")
    epochs = int(os.environ.get("CONV_EPOCHS", "40"))
    mpnn = os.environ.get("CONV_MODEL", "MACE")
    n_train, n_val = 512, 128

    torch.manual_seed(11)
    if mpnn == "MACE":
        model = B.build_model(device, precision=precision, seed=11)
    else:
        from hydragnn_amd.models.create import (create_model,
                                                resolve_precision)
        extra = {}
        if mpnn == "SchNet":
            extra = dict(num_gaussians=32, num_filters=64)
        model = create_model(
            mpnn_type=mpnn, input_dim=1, hidden_dim=64,
            output_dim=[1], output_type=["node"],
            output_heads=B.MODEL_CONFIG["output_heads"],
            activation_function="silu", loss_function_type="mse",
            task_weights=[1.0], num_conv_layers=3, num_nodes=21,
            max_neighbours=30, radius=7.0,
            enable_interatomic_potential=True, energy_weight=1.0,
            energy_peratom_weight=1.0, force_weight=100.0,
            equivariance=True, use_gpu=False, **extra)
        _, pd, _ = resolve_precision(precision)
        model = model.to(device=device, dtype=pd)
    default_lr = {"MACE": 2e-3, "SchNet": 2e-3, "EGNN": 1e-3,
                  "PAINN": 3e-4}.get(mpnn, 1e-3)
    lr = float(os.environ.get("CONV_LR", default_lr))
    opt = torch.optim.AdamW(model.parameters(), lr=lr, foreach=True)
    clip = os.environ.get("CONV_CLIP")
    if clip is None and mpnn == "EGNN":
        clip = "10.0"   # EGNN diverges unclipped at useful lr
    if clip is not None:
        opt._hydragnn_grad_clip = float(clip)
    sched = torch.optim.lr_scheduler.CosineAnnealingLR(opt, epochs)

    # tame LJ landscape (min pair distance 0.95 sigma) so the
    # regression targets are O(1-100), not 1e6
    ds = md17_shape_dataset_fast(n_train + n_val, seed=4,
                                 min_dist=0.95, max_push=40)
    train_ds, val_ds = ds[:n_train], ds[n_train:]
    train_loader = DataLoader(train_ds, batch_size=32, shuffle=True,
                              collate_fn=_collate)

    def val_metrics():
        model.eval()
        e_ae = f_ae = 0.0
        n_g = n_a = 0
        from hydragnn_amd.data import Batch
        for i in range(0, n_val, 32):
            b = Batch.from_data_list(
                [d.clone() for d in val_ds[i:i + 32]]).to(device)
            b.pos.requires_grad_(True)
            pred = model(b)
            e = pred[0].float()
            from hydragnn_amd.ops import scatter
            e_g = scatter(e, b.batch, b.num_graphs, "sum").squeeze(-1)
            f = -torch.autograd.grad(e_g.sum(), b.pos)[0].float()
            e_ae += (e_g - b.energy.squeeze().float()).abs().sum().item()
            f_ae += (f - b.forces.float()).abs().sum().item()
            n_g += b.num_graphs
            n_a += b.pos.shape[0] * 3
        model.train()
        return e_ae / n_g, f_ae / n_a

    traj = []
    t0 = time.time()
    for ep in range(epochs):
        err, tasks = train(train_loader, model, opt, 0,
                           precision=precision)
        sched.step()
        e_mae, f_mae = val_metrics()
        rec = {"epoch": ep, "train_loss": float(err),
               "val_energy_mae": e_mae, "val_force_mae": f_mae,
               "elapsed_s": round(time.time() - t0, 1)}
        traj.append(rec)
        print(json.dumps(rec), flush=True)

    out = {
        "config": f"{mpnn}-MLIP {precision}, MD17-shape synthetic (LJ), "
                  "512 train / 128 val, batch 32, AdamW 2e-3 cosine",
        "device": torch.cuda.get_device_name(0) if device != "cpu"
                  else "cpu",
        "precision": precision,
        "trajectory": traj,
        "loss_drop": traj[0]["train_loss"] / max(traj[-1]["train_loss"],
                                                 1e-12),
        "force_mae_drop": traj[0]["val_force_mae"]
                          / max(traj[-1]["val_force_mae"], 1e-12),
    }
    os.makedirs("gpurun_out", exist_ok=True)
    out_name = (f"gpurun_out/convergence_{mpnn.lower()}"
                "_md17.json")
    with open(out_name, "w") as f:
        json.dump(out, f, indent=1)
    print(f"loss drop x{out['loss_drop']:.1f}, "
          f"force MAE drop x{out['force_mae_drop']:.1f}")


if __name__ == "__main__":
    main()
