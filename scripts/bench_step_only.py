"""Step-only (captured-kernel) benchmark: times the resident-batch
hipGraph replay path (bench.build_model_and_batch) WITHOUT the
dataloader/H2D — the upper bound the train-loop bench (bench.py) is
measured against.  VERDICT r1 asked for the two side by side."""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch", type=int, default=1024)
    args = ap.parse_args()

    from bench import build_model_and_batch
    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    model, batch, step = build_model_and_batch(device=device,
                                               local_batch=args.batch)
    for _ in range(args.warmup):
        step()
    if device != "cpu":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if device != "cpu":
        torch.cuda.synchronize()
    t1 = time.perf_counter()
    el = t1 - t0
    print(json.dumps({
        "metric": "graphs/sec captured-step (MACE, MD17-shape)",
        "value": args.batch * args.steps / el,
        "ms_per_step": el / args.steps * 1000.0,
        "steps": args.steps, "warmup": args.warmup,
        "batch": args.batch,
    }))


if __name__ == "__main__":
    main()
