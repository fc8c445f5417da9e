#!/usr/bin/env python3
"""Per-shape conv microbenchmark: fda conv_igemm vs the library (MIOpen).

Times fwd and dgrad for every ResNet-34 bs-96 body shape; prints a table
with effective TFLOP/s. Run on a GPU box:
    python tools/bench_conv.py [--batch 96] [--iters 50]
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fluxdistributed_amd.ops.native import require_native  # noqa: E402

# (C, H, W, K, R, stride, count_in_resnet34)
SHAPES = [
    (64, 56, 56, 64, 3, 1, 7),      # layer1 3x3 (6 full + conv2 of first)
    (64, 56, 56, 128, 3, 2, 1),     # layer2 entry
    (64, 56, 56, 128, 1, 2, 1),     # layer2 downsample
    (128, 28, 28, 128, 3, 1, 7),
    (128, 28, 28, 256, 3, 2, 1),
    (128, 28, 28, 256, 1, 2, 1),
    (256, 14, 14, 256, 3, 1, 11),
    (256, 14, 14, 512, 3, 2, 1),
    (256, 14, 14, 512, 1, 2, 1),
    (512, 7, 7, 512, 3, 1, 5),
]


def timeit(fn, iters, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=96)
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--mode", choices=["fwd", "dgrad", "both"], default="both")
    p.add_argument("--only", type=int, default=-1,
                   help="run only SHAPES[i] (for PMC profiling runs)")
    p.add_argument("--fda-only", action="store_true",
                   help="skip the library arms (clean kernel-trace)")
    args = p.parse_args()
    assert torch.cuda.is_available()
    torch.backends.cudnn.benchmark = True
    C_ = require_native("conv_igemm_fwd")
    n = args.batch

    print(f"# batch={n}  times in us, eff TF = 2*M*N*K / t")
    print(f"{'shape':34s} {'fda_fwd':>9s} {'lib_fwd':>9s} {'fda_TF':>7s} {'lib_TF':>7s}"
          f" | {'fda_dg':>9s} {'lib_dg':>9s} | {'fda_wg':>9s} {'lib_wg':>9s}")
    tot_fda_f = tot_lib_f = tot_fda_d = tot_lib_d = 0.0
    tot_fda_w = tot_lib_w = 0.0
    shapes = SHAPES if args.only < 0 else [SHAPES[args.only]]
    for (c, h, w, k, r, s, cnt) in shapes:
        pad = r // 2
        P, Q = (h + 2 * pad - r) // s + 1, (w + 2 * pad - r) // s + 1
        x = torch.randn(n, c, h, w, device="cuda").bfloat16() \
            .contiguous(memory_format=torch.channels_last)
        wt = torch.randn(k, c, r, r, device="cuda").bfloat16() \
            .contiguous(memory_format=torch.channels_last)
        flops = 2.0 * n * P * Q * k * c * r * r

        t_ff = timeit(lambda: C_.conv_igemm_fwd(x, wt, s, s, pad, pad), args.iters)
        t_lf = (timeit(lambda: torch.nn.functional.conv2d(x, wt, None, s, pad),
                       args.iters) if not args.fda_only else float("inf"))
        gy = torch.randn(n, k, P, Q, device="cuda").bfloat16() \
            .contiguous(memory_format=torch.channels_last)
        wtt = wt.permute(2, 3, 1, 0).reshape(r * r * c, k).contiguous()
        t_fd = timeit(lambda: C_.conv_igemm_dgrad(gy, wtt, c, h, w, r, r,
                                                  s, s, pad, pad), args.iters)
        xg = x.requires_grad_(True)

        def lib_dgrad():
            return torch.ops.aten.convolution_backward(
                gy, xg, wt, None, [s, s], [pad, pad], [1, 1], False, [0, 0],
                1, [True, False, False])[0]
        t_ld = timeit(lib_dgrad, args.iters) if not args.fda_only else float("inf")
        t_fw = timeit(lambda: C_.conv_igemm_wgrad(gy, x, r, r, s, s, pad, pad),
                      args.iters)

        def lib_wgrad():
            return torch.ops.aten.convolution_backward(
                gy, xg, wt, None, [s, s], [pad, pad], [1, 1], False, [0, 0],
                1, [False, True, False])[1]
        t_lw = timeit(lib_wgrad, args.iters) if not args.fda_only else float("inf")

        name = f"{c}x{h}x{w} k{k} {r}x{r} s{s} x{cnt}"
        print(f"{name:34s} {t_ff*1e6:9.1f} {t_lf*1e6:9.1f} "
              f"{flops/t_ff/1e12:7.1f} {flops/t_lf/1e12:7.1f} | "
              f"{t_fd*1e6:9.1f} {t_ld*1e6:9.1f} | "
              f"{t_fw*1e6:9.1f} {t_lw*1e6:9.1f}")
        tot_fda_f += t_ff * cnt
        tot_lib_f += t_lf * cnt
        tot_fda_d += t_fd * cnt
        tot_lib_d += t_ld * cnt
        tot_fda_w += t_fw * cnt
        tot_lib_w += t_lw * cnt
    print(f"{'TOTAL (weighted by layer count)':34s} {tot_fda_f*1e6:9.1f} "
          f"{tot_lib_f*1e6:9.1f} {'':7s} {'':7s} | {tot_fda_d*1e6:9.1f} "
          f"{tot_lib_d*1e6:9.1f} | {tot_fda_w*1e6:9.1f} {tot_lib_w*1e6:9.1f}")


if __name__ == "__main__":
    main()
