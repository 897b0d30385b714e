"""Multidataset + DeepSpeed example (reference
examples/multidataset_deepspeed): wraps the multidataset trainer in
deepspeed.initialize (ZeRO-1 + bf16 via parse_deepspeed_config).
Falls back to DDP with a notice when deepspeed is not installed."""
import argparse
import os
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, os.path.join(HERE, ".."))
sys.path.insert(0, os.path.join(HERE, "..", ".."))

import torch  # noqa: E402

from hydragnn_amd.preprocess import create_dataloaders  # noqa: E402
from hydragnn_amd.models import create_model_config  # noqa: E402
from hydragnn_amd.train import train as train_fn  # noqa: E402
from hydragnn_amd.utils.config import update_config  # noqa: E402
from hydragnn_amd.utils.distributed import (  # noqa: E402
    deepspeed_model_wrapper, distributed_model_wrapper, setup_ddp)
from hydragnn_amd.utils.optimizer import select_optimizer  # noqa: E402

sys.path.insert(0, os.path.join(HERE, "..", "multidataset"))
from train import CONFIG, make_dataset  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--num_epoch", type=int, default=3)
    args = p.parse_args()
    setup_ddp()
    torch.manual_seed(3)
    dataset = make_dataset("dsA", 96, 1) + make_dataset("dsB", 48, 2)
    config = dict(CONFIG)
    config["NeuralNetwork"]["Training"]["deepspeed"] = {
        "zero_stage": 1, "bf16": True}
    loaders = create_dataloaders(dataset, dataset, dataset, 16,
                                 config=config)
    config = update_config(config, *loaders)
    model = create_model_config(config["NeuralNetwork"], use_gpu=False)
    opt = select_optimizer(
        model, config["NeuralNetwork"]["Training"]["Optimizer"])
    try:
        model, opt = deepspeed_model_wrapper(model, opt, config)
        print("deepspeed engine active")
    except ImportError as e:
        print(f"deepspeed unavailable ({e}); DDP fallback")
        model = distributed_model_wrapper(model)
    for epoch in range(args.num_epoch):
        err, _ = train_fn(loaders[0], model, opt, 0)
        print(f"epoch {epoch} loss {float(err):.6f}")


if __name__ == "__main__":
    main()
