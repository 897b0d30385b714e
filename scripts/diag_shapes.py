"""torch.profiler over a few eager steps: find remaining degenerate
GEMMs (aten::mm/addmm/bmm with tiny M) by input shape."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch  # noqa: E402


def main():
    os.environ["HYDRAGNN_CAPTURE"] = "0"
    import bench as B
    from hydragnn_amd.train import train
    from hydragnn_amd.ops.fused_adamw import FusedAdamW

    model = B.build_model("cuda:0")
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    warm, timed = B.make_loaders(0, 2, 1, 1024, True)
    train(warm, model, opt, 0, precision="bf16_pure")
    torch.cuda.synchronize()
    with torch.profiler.profile(
            activities=[torch.profiler.ProfilerActivity.CPU,
                        torch.profiler.ProfilerActivity.CUDA],
            record_shapes=True) as prof:
        train(timed, model, opt, 0, precision="bf16_pure")
        torch.cuda.synchronize()
    evs = prof.key_averages(group_by_input_shape=True)
    import os as _os
    kinds = _os.environ.get("DIAG_OPS", "mm,bmm,addmv,addmm,matmul,"
                            "linear").split(",")
    rows = [(e.device_time_total, e.key, e.input_shapes, e.count)
            for e in evs
            if any(k in e.key for k in kinds)]
    rows.sort(reverse=True)
    for t, k, shp, c in rows[:25]:
        print(f"{t/1000:9.2f} ms x{c:<4} {k:<18} {shp}")


if __name__ == "__main__":
    main()
