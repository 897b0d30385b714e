"""Micro-bench etp_reduce (the B-slot channel-reduce gradient):
achieved bandwidth vs the HBM roofline, across block sizes."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch  # noqa: E402


def run(block):
    os.environ["HYDRAGNN_ETP_BLOCK"] = str(block)
    from hydragnn_amd.models.mace.blocks import EdgeTensorProduct
    from hydragnn_amd.ops import etp as etp_mod

    tp = EdgeTensorProduct(2, 2, 2)  # lmax config of the headline
    table = tp.etp_table
    da, db, dg, do = table.dims
    E, C = 332358, 64
    dev = "cuda:0"
    A = torch.randn(E, C, da, device=dev).bfloat16()
    Cw = torch.randn(E, C, dg, device=dev).bfloat16()
    D = torch.randn(E, C, do, device=dev).bfloat16()
    ext = etp_mod.get_extension(required=True)
    ent, coefs, _ = table.device_tensors(dev)

    def call():
        return ext.etp_reduce(A, Cw, D, ent, coefs, db)

    for _ in range(3):
        call()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 20
    for _ in range(iters):
        call()
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    gb = (A.numel() + Cw.numel() + D.numel()) * 2 / 1e9
    print(f"block {block}: {us:8.1f} us  "
          f"{gb / (us / 1e6):7.1f} GB/s  (read {gb:.2f} GB/call, "
          f"dims da={da} db={db} dg={dg} do={do} "
          f"n_ent={table.entries.shape[0]})")


if __name__ == "__main__":
    run(int(sys.argv[1]) if len(sys.argv) > 1 else 256)
