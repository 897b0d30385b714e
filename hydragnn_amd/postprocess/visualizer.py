"""Matplotlib visualization suite (reference:
hydragnn/postprocess/visualizer.py, 740 LoC): predicted-vs-true scatter
per head, training history curves, error histograms. Thin but
API-compatible; degrades to no-op without matplotlib."""

from __future__ import annotations

import os
from typing import List, Optional

import torch

try:
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    _HAS_MPL = True
except ImportError:  # pragma: no cover
    _HAS_MPL = False


class Visualizer:
    def __init__(self, model_with_config_name: str,
                 node_feature: Optional[List] = None,
                 num_heads: int = 1, head_dims: Optional[List[int]] = None,
                 path: str = "./logs/"):
        self.name = model_with_config_name
        self.num_heads = num_heads
        self.head_dims = head_dims or [1] * num_heads
        self.outdir = os.path.join(path, model_with_config_name)
        os.makedirs(self.outdir, exist_ok=True)
        self.history = {"train": [], "val": [], "test": []}

    def add_history(self, train_err: float, val_err: float,
                    test_err: float):
        self.history["train"].append(train_err)
        self.history["val"].append(val_err)
        self.history["test"].append(test_err)

    def plot_history(self):
        if not _HAS_MPL or not self.history["train"]:
            return
        fig, ax = plt.subplots(figsize=(6, 4))
        for k, v in self.history.items():
            ax.plot(v, label=k)
        ax.set_xlabel("epoch")
        ax.set_ylabel("loss")
        ax.set_yscale("log")
        ax.legend()
        fig.tight_layout()
        fig.savefig(os.path.join(self.outdir, "history.png"), dpi=120)
        plt.close(fig)

    def create_scatter_plots(self, true_values, predicted_values,
                             output_names=None, iepoch=None):
        if not _HAS_MPL:
            return
        n = len(true_values)
        fig, axes = plt.subplots(1, max(n, 1), figsize=(5 * max(n, 1), 4))
        if n == 1:
            axes = [axes]
        for ihead in range(n):
            t = true_values[ihead].detach().cpu().flatten()
            p = predicted_values[ihead].detach().cpu().flatten()
            if t.numel() == 0:
                continue
            ax = axes[ihead]
            ax.scatter(t.numpy(), p.numpy(), s=4, alpha=0.5)
            lo, hi = float(t.min()), float(t.max())
            ax.plot([lo, hi], [lo, hi], "k--", lw=1)
            name = (output_names[ihead] if output_names else
                    f"head {ihead}")
            rmse = float(torch.sqrt(((t - p) ** 2).mean()))
            ax.set_title(f"{name} (RMSE {rmse:.4f})")
            ax.set_xlabel("true")
            ax.set_ylabel("predicted")
        fig.tight_layout()
        suffix = f"_epoch{iepoch}" if iepoch is not None else ""
        fig.savefig(os.path.join(self.outdir, f"scatter{suffix}.png"),
                    dpi=120)
        plt.close(fig)

    def create_error_histograms(self, true_values, predicted_values,
                                output_names=None):
        if not _HAS_MPL:
            return
        n = len(true_values)
        fig, axes = plt.subplots(1, max(n, 1), figsize=(5 * max(n, 1), 4))
        if n == 1:
            axes = [axes]
        for ihead in range(n):
            t = true_values[ihead].detach().cpu().flatten()
            p = predicted_values[ihead].detach().cpu().flatten()
            if t.numel() == 0:
                continue
            axes[ihead].hist((p - t).numpy(), bins=40)
            axes[ihead].set_xlabel("error")
        fig.tight_layout()
        fig.savefig(os.path.join(self.outdir, "error_hist.png"), dpi=120)
        plt.close(fig)
