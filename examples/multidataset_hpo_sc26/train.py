"""SC26-variant multidataset HPO (reference
examples/multidataset_hpo_sc26): adds the at-scale knobs — node-budget
cost-aware batching, HYDRAGNN_MAX_NUM_BATCH clamp, MPI metric
aggregation backend — around the multidataset trainer."""
import argparse
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, os.path.join(HERE, "..", ".."))

from hydragnn_amd.utils.hpo import run_search  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--trials", type=int, default=2)
    p.add_argument("--num_epoch", type=int, default=2)
    args = p.parse_args()

    def objective(cfg):
        env = dict(os.environ, MASTER_ADDR="127.0.0.1",
                   HYDRAGNN_HPO_LR=str(cfg["lr"]),
                   HYDRAGNN_MAX_NUM_BATCH="20",
                   HYDRAGNN_AGGR_BACKEND="mpi")
        r = subprocess.run(
            [sys.executable,
             os.path.join(HERE, "..", "multidataset", "train.py"),
             "--num_epoch", str(args.num_epoch)],
            capture_output=True, text=True, timeout=600, env=env)
        if r.returncode != 0:
            return 1e9
        for line in reversed(r.stdout.splitlines()):
            if "loss" in line:
                try:
                    return float(line.rsplit(" ", 1)[-1])
                except ValueError:
                    continue
        return 1e9

    best_cfg, best_val = run_search(
        objective, {"lr": (1e-4, 1e-2, "log")},
        num_trials=args.trials, seed=26, use_deephyper=True)
    print(f"best params {best_cfg} -> {best_val}")


if __name__ == "__main__":
    main()
