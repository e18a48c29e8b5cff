"""Calibrate achievable HBM streaming bandwidth on this box.

Times torch's own kernels (copy, sum, add) plus our bn_stats/bn_bwd_reduce
on the ResNet stem shape, so kernel efficiency is judged against what the
machine actually delivers, not the paper peak.
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def t(fn, iters=20, warm=3):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    torch.cuda.set_device(0)
    M, C = 64 * 112 * 112, 64
    n = M * C
    x = torch.randn(n, device="cuda").to(torch.bfloat16)
    y = torch.empty_like(x)
    xf = torch.randn(n // 2, device="cuda")
    yf = torch.empty_like(xf)
    gb = n * 2 / 1e9

    def show(k, sec, bytes_gb):
        print("%-22s %8.1f us  %6.2f TB/s" % (k, sec * 1e6,
                                              bytes_gb / sec / 1e3),
              flush=True)

    show("copy_bf16 (R+W)", t(lambda: y.copy_(x)), 2 * gb)
    show("copy_f32 (R+W)", t(lambda: yf.copy_(xf)), 2 * gb)
    show("sum_bf16 (R)", t(lambda: x.sum()), gb)
    show("sum_f32 (R)", t(lambda: xf.sum()), gb)
    show("add_bf16 (2R+W)", t(lambda: torch.add(x, y, out=y)), 3 * gb)
    show("sum_2d_ch (R)", t(lambda: x.view(M, C).sum(0)), gb)

    from kungfu_amd import _hip

    s = torch.cuda.current_stream().cuda_stream
    sums = torch.zeros(16 * C, dtype=torch.float32, device="cuda")
    mean = torch.zeros(C, dtype=torch.float32, device="cuda")
    rstd = torch.ones(C, dtype=torch.float32, device="cuda")
    mask = torch.full((M * (C // 8),), 255, dtype=torch.uint8,
                      device="cuda")
    show("bn_stats (R)",
         t(lambda: _hip.bn_stats(x.data_ptr(), M, C, sums.data_ptr(), s)),
         gb)
    dy = y
    show("bn_bwd_reduce (2R)",
         t(lambda: _hip.bn_bwd_reduce(dy.data_ptr(), x.data_ptr(),
                                      mask.data_ptr(), mean.data_ptr(),
                                      rstd.data_ptr(), M, C,
                                      sums.data_ptr(), s)), 2 * gb)


if __name__ == "__main__":
    main()
