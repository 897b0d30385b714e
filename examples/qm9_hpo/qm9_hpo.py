"""QM9 hyper-parameter optimization example (reference
examples/qm9_hpo/qm9_hpo.py pattern): random search over architecture
hyper-parameters, each trial a short full training run scored on
validation loss.  DeepHyper is not in this image; the offline
`run_random_search` helper (utils/hpo/deephyper.py) provides the same
loop, and `parse_slurm_nodelist` maps trials onto SLURM allocations
the way the reference's deephyper launcher does.

Run: python examples/qm9_hpo/qm9_hpo.py --trials 5 --epochs 4
"""

import argparse
import copy
import json
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from hydragnn_amd.preprocess import create_dataloaders, split_dataset  # noqa: E402
from hydragnn_amd.models import create_model_config  # noqa: E402
from hydragnn_amd.train import train_validate_test  # noqa: E402
from hydragnn_amd.utils.config import update_config  # noqa: E402
from hydragnn_amd.utils.distributed import (  # noqa: E402
    setup_ddp, distributed_model_wrapper)
from hydragnn_amd.utils.hpo.deephyper import run_random_search  # noqa: E402
from hydragnn_amd.utils.optimizer import select_optimizer  # noqa: E402

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "qm9"))
from qm9 import build_dataset  # noqa: E402


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--trials", type=int, default=5)
    parser.add_argument("--epochs", type=int, default=4)
    parser.add_argument("--samples", type=int, default=120)
    args = parser.parse_args()

    setup_ddp()
    with open(os.path.join(os.path.dirname(__file__), "..", "qm9",
                           "qm9.json")) as f:
        base_config = json.load(f)
    dataset = build_dataset(num_samples=args.samples)
    splits = split_dataset(dataset, 0.8, False)

    def objective(params):
        config = copy.deepcopy(base_config)
        arch = config["NeuralNetwork"]["Architecture"]
        arch["hidden_dim"] = params["hidden_dim"]
        arch["num_conv_layers"] = params["num_conv_layers"]
        arch["mpnn_type"] = params["mpnn_type"]
        config["NeuralNetwork"]["Training"]["num_epoch"] = args.epochs
        config["NeuralNetwork"]["Training"]["Optimizer"][
            "learning_rate"] = params["lr"]
        loaders = create_dataloaders(*splits, config["NeuralNetwork"]
                                     ["Training"]["batch_size"],
                                     config=config)
        trial_cfg = update_config(config, *loaders)
        model = create_model_config(trial_cfg["NeuralNetwork"],
                                    verbosity=0)
        model = distributed_model_wrapper(model, verbosity=0)
        optimizer = select_optimizer(model, trial_cfg["NeuralNetwork"]
                                     ["Training"])
        scheduler = torch.optim.lr_scheduler.ReduceLROnPlateau(
            optimizer, patience=2)
        train_validate_test(model, optimizer, *loaders, None, scheduler,
                            trial_cfg["NeuralNetwork"], "qm9_hpo_trial",
                            0, create_plots=False)
        core = model.module if hasattr(model, "module") else model
        core.eval()
        with torch.no_grad():
            total, count = 0.0, 0
            from hydragnn_amd.train import get_head_indices
            for batch in loaders[1]:
                pred = core(batch)
                loss, _ = core.loss(pred, batch.y,
                                    get_head_indices(core, batch))
                total += float(loss)
                count += 1
        return total / max(count, 1)

    space = {
        "hidden_dim": [16, 32, 64],
        "num_conv_layers": [2, 3],
        "mpnn_type": ["GIN", "SAGE", "EGNN"],
        "lr": (1e-4, 1e-2, "log"),
    }
    best, val, history = run_random_search(objective, space,
                                           num_trials=args.trials,
                                           seed=17)
    print("best params:", best)
    print("best val loss:", val)
    for i, (p, v) in enumerate(history):
        print(f"trial {i}: {v:.5f} {p}")


if __name__ == "__main__":
    main()
