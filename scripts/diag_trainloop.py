"""Diagnose train-loop bench: per-step fetch vs step time, capture
engagement, worker count."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch  # noqa: E402


def main():
    import bench as B
    from hydragnn_amd.train import train
    from hydragnn_amd.train.train_validate_test import (
        _compute_loss, get_autocast_and_scaler)
    from hydragnn_amd.train.captured import get_or_build_stepper
    from hydragnn_amd.models.create import resolve_precision
    from hydragnn_amd.utils.distributed import distributed_model_wrapper

    device = "cuda:0"
    batch_sz = int(os.environ.get("DIAG_BATCH", "1024"))
    steps = int(os.environ.get("DIAG_STEPS", "12"))
    model = B.build_model(device)
    model = distributed_model_wrapper(model)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3, foreach=True)
    warm, timed = B.make_loaders(0, steps, 2, batch_sz, True)
    print("workers:", timed.num_workers, "cpus:", os.cpu_count())

    t0 = time.time()
    train(warm, model, opt, 0, precision="bf16")
    torch.cuda.synchronize()
    print(f"warmup epoch (2 batches + capture): {time.time()-t0:.2f}s")
    base = model.module if hasattr(model, "module") else model
    stepper = getattr(base, "_hip_captured_step", None)
    print("stepper engaged:", stepper not in (None, False))

    for epoch in range(2):
        it = iter(timed)
        tf_sum = ts_sum = 0.0
        for i in range(steps):
            t0 = time.time()
            data = next(it)
            t1 = time.time()
            assert stepper and stepper.matches(data)
            stepper._copy_in(data)
            torch.cuda.synchronize()
            t2 = time.time()
            stepper.graph.replay()
            torch.cuda.synchronize()
            t3 = time.time()
            stepper.grad_sync()
            stepper.opt.step()
            torch.cuda.synchronize()
            t4 = time.time()
            tf_sum += t1 - t0
            ts_sum += t4 - t1
            print(f"e{epoch} step {i}: fetch {1000*(t1-t0):6.1f}  "
                  f"copy {1000*(t2-t1):6.1f}  "
                  f"replay {1000*(t3-t2):6.1f}  "
                  f"opt {1000*(t4-t3):6.1f} ms")
        print(f"e{epoch}: mean fetch {1000*tf_sum/steps:.1f} ms, "
              f"mean step {1000*ts_sum/steps:.1f} ms")


if __name__ == "__main__":
    main()
