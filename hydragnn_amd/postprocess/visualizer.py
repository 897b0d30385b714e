"""Matplotlib visualization suite (reference:
hydragnn/postprocess/visualizer.py, 740 LoC): predicted-vs-true scatter
per head, training history curves, error histograms, 2D-density parity
analysis, per-size error breakdowns, vector-component parity and
graph-size statistics; degrades to no-op without matplotlib."""

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

    @staticmethod
    def add_identity(axes, *line_args, **line_kwargs):
        """y = x guide that tracks axis-limit changes (reference
        visualizer.py:612)."""
        (identity,) = axes.plot([], [], *line_args, **line_kwargs)

        def callback(ax):
            lo = max(ax.get_xlim()[0], ax.get_ylim()[0])
            hi = min(ax.get_xlim()[1], ax.get_ylim()[1])
            identity.set_data([lo, hi], [lo, hi])

        callback(axes)
        axes.callbacks.connect("xlim_changed", callback)
        axes.callbacks.connect("ylim_changed", callback)
        return axes

    def create_plot_global_analysis(self, true_values, predicted_values,
                                    output_names=None, iepoch=None):
        """2D-density parity + conditional mean-|error| per head
        (reference create_plot_global_analysis pattern)."""
        if not _HAS_MPL:
            return
        import numpy as np
        n = len(true_values)
        fig, axes = plt.subplots(2, max(n, 1),
                                 figsize=(5 * max(n, 1), 8),
                                 squeeze=False)
        for ihead in range(n):
            t = true_values[ihead].detach().cpu().flatten().numpy()
            p = predicted_values[ihead].detach().cpu().flatten().numpy()
            if t.size == 0:
                continue
            ax = axes[0][ihead]
            h, xe, ye = np.histogram2d(t, p, bins=40)
            xc, yc = 0.5 * (xe[:-1] + xe[1:]), 0.5 * (ye[:-1] + ye[1:])
            if h.max() > 0:
                ax.contourf(xc, yc, h.T + 1e-12, levels=12,
                            cmap="viridis")
            self.add_identity(ax, "w--", lw=1)
            ax.set_xlabel("true")
            ax.set_ylabel("predicted")
            # conditional mean |error| over true-value bins
            ax2 = axes[1][ihead]
            bins = np.linspace(t.min(), t.max() + 1e-12, 21)
            which = np.digitize(t, bins) - 1
            err = np.abs(p - t)
            cm = [err[which == b].mean() if (which == b).any() else 0.0
                  for b in range(20)]
            ax2.plot(0.5 * (bins[:-1] + bins[1:]), cm, "o-")
            ax2.set_xlabel("true")
            ax2.set_ylabel("mean |error|")
        fig.tight_layout()
        suffix = f"_epoch{iepoch}" if iepoch is not None else ""
        fig.savefig(os.path.join(self.outdir,
                                 f"global_analysis{suffix}.png"), dpi=120)
        plt.close(fig)

    def create_error_histogram_per_node(self, true_values,
                                        predicted_values, node_counts,
                                        output_names=None):
        """Error distribution grouped by graph size (reference
        create_error_histogram_per_node pattern)."""
        if not _HAS_MPL:
            return
        import numpy as np
        t = true_values[0].detach().cpu().flatten().numpy()
        p = predicted_values[0].detach().cpu().flatten().numpy()
        nc = torch.as_tensor(node_counts).cpu().flatten().numpy()
        if t.size == 0 or nc.size != t.size:
            return
        fig, ax = plt.subplots(figsize=(6, 4))
        sizes = np.unique(nc)
        for s in sizes[:8]:
            sel = nc == s
            ax.hist((p - t)[sel], bins=30, histtype="step",
                    label=f"N={int(s)}")
        ax.set_xlabel("error")
        ax.legend(fontsize=7)
        fig.tight_layout()
        fig.savefig(os.path.join(self.outdir, "error_hist_per_size.png"),
                    dpi=120)
        plt.close(fig)

    def create_parity_plot_vector(self, true_values, predicted_values,
                                  components=("x", "y", "z"),
                                  name="forces"):
        """Per-component parity for vector heads (reference
        create_parity_plot_vector pattern)."""
        if not _HAS_MPL:
            return
        t = true_values.detach().cpu().reshape(-1, len(components))
        p = predicted_values.detach().cpu().reshape(-1, len(components))
        fig, axes = plt.subplots(1, len(components),
                                 figsize=(5 * len(components), 4))
        for i, c in enumerate(components):
            ax = axes[i] if len(components) > 1 else axes
            ax.scatter(t[:, i].numpy(), p[:, i].numpy(), s=3, alpha=0.4)
            self.add_identity(ax, "k--", lw=1)
            rmse = float(torch.sqrt(((t[:, i] - p[:, i]) ** 2).mean()))
            ax.set_title(f"{name}.{c} (RMSE {rmse:.4f})")
            ax.set_xlabel("true")
            ax.set_ylabel("predicted")
        fig.tight_layout()
        fig.savefig(os.path.join(self.outdir, f"parity_{name}.png"),
                    dpi=120)
        plt.close(fig)

    def num_nodes_plot(self, node_counts):
        """Graph-size histogram of the dataset (reference
        num_nodes_plot)."""
        if not _HAS_MPL:
            return
        nc = torch.as_tensor(node_counts).cpu().flatten().numpy()
        fig, ax = plt.subplots(figsize=(5, 4))
        ax.hist(nc, bins=min(40, max(len(set(nc.tolist())), 2)))
        ax.set_xlabel("nodes per graph")
        ax.set_ylabel("count")
        fig.tight_layout()
        fig.savefig(os.path.join(self.outdir, "num_nodes.png"), dpi=120)
        plt.close(fig)

    def create_plot_global(self, true_values, predicted_values,
                           output_names=None):
        """One parity panel per head in a single figure (reference
        create_plot_global pattern)."""
        if not _HAS_MPL:
            return
        n = len(true_values)
        fig, axes = plt.subplots(1, max(n, 1),
                                 figsize=(5 * max(n, 1), 4),
                                 squeeze=False)
        for ihead in range(n):
            ax = axes[0][ihead]
            t = true_values[ihead].detach().cpu().flatten().numpy()
            p = predicted_values[ihead].detach().cpu().flatten().numpy()
            ax.scatter(t, p, s=4, alpha=0.5)
            self.add_identity(ax, "k--", lw=1)
            name = (output_names[ihead] if output_names
                    and ihead < len(output_names) else f"head{ihead}")
            ax.set_title(name)
            ax.set_xlabel("true")
            ax.set_ylabel("predicted")
        fig.tight_layout()
        fig.savefig(os.path.join(self.outdir, "parity_global.png"),
                    dpi=120)
        plt.close(fig)

    def create_parity_plot_and_error_histogram_scalar(
            self, varname, true_values, predicted_values, iepoch=None):
        """Scalar head: parity scatter + error PDF side by side; for
        per-node-slot outputs (fixed-size graphs) a grid of per-slot
        parity panels plus SUM panels (reference
        create_parity_plot_and_error_histogram_scalar pattern)."""
        if not _HAS_MPL:
            return
        import numpy as np
        t = true_values.detach().cpu().numpy()
        p = predicted_values.detach().cpu().numpy()
        if t.ndim == 1 or t.shape[-1] == 1:
            t, p = t.reshape(-1), p.reshape(-1)
            fig, axes = plt.subplots(1, 2, figsize=(10, 4.5))
            axes[0].scatter(t, p, s=4, alpha=0.5)
            self.add_identity(axes[0], "k--", lw=1)
            axes[0].set_title(varname)
            axes[0].set_xlabel("true")
            axes[0].set_ylabel("predicted")
            hist, edges = np.histogram(p - t, bins=40, density=True)
            axes[1].plot(0.5 * (edges[:-1] + edges[1:]), hist, "ro")
            axes[1].set_title(f"{varname}: error PDF")
        else:
            nslots = t.shape[1]
            ncol = int(np.ceil(np.sqrt(nslots + 1)))
            nrow = int(np.ceil((nslots + 1) / ncol))
            fig, axes = plt.subplots(nrow, ncol,
                                     figsize=(3 * ncol, 3 * nrow),
                                     squeeze=False)
            flat = axes.flatten()
            for s in range(nslots):
                flat[s].scatter(t[:, s], p[:, s], s=4, alpha=0.5)
                self.add_identity(flat[s], "k--", lw=1)
                flat[s].set_title(f"slot {s}", fontsize=8)
            flat[nslots].scatter(t.sum(1), p.sum(1), s=10)
            self.add_identity(flat[nslots], "k--", lw=1)
            flat[nslots].set_title("SUM", fontsize=8)
            for s in range(nslots + 1, len(flat)):
                flat[s].axis("off")
        fig.tight_layout()
        suffix = f"_epoch{iepoch}" if iepoch is not None else ""
        fig.savefig(os.path.join(
            self.outdir, f"parity_hist_{varname}{suffix}.png"), dpi=120)
        plt.close(fig)

    def create_parity_plot_per_node_vector(
            self, varname, true_values, predicted_values,
            num_nodes, iepoch=None):
        """Vector node targets on fixed-size graphs: per-node-slot
        parity of the vector magnitude (reference
        create_parity_plot_per_node_vector pattern)."""
        if not _HAS_MPL:
            return
        import numpy as np
        t = true_values.detach().cpu().reshape(-1, num_nodes, 3).numpy()
        p = (predicted_values.detach().cpu()
             .reshape(-1, num_nodes, 3).numpy())
        tm = np.linalg.norm(t, axis=-1)
        pm = np.linalg.norm(p, axis=-1)
        ncol = int(np.ceil(np.sqrt(num_nodes)))
        nrow = int(np.ceil(num_nodes / ncol))
        fig, axes = plt.subplots(nrow, ncol,
                                 figsize=(3 * ncol, 3 * nrow),
                                 squeeze=False)
        flat = axes.flatten()
        for s in range(num_nodes):
            flat[s].scatter(tm[:, s], pm[:, s], s=4, alpha=0.5)
            self.add_identity(flat[s], "k--", lw=1)
            flat[s].set_title(f"node {s}", fontsize=8)
        for s in range(num_nodes, len(flat)):
            flat[s].axis("off")
        fig.tight_layout()
        suffix = f"_epoch{iepoch}" if iepoch is not None else ""
        fig.savefig(os.path.join(
            self.outdir, f"parity_pernode_{varname}{suffix}.png"),
            dpi=120)
        plt.close(fig)
