#!/usr/bin/env python3
"""Per-shape conv timings: native igemm (fwd/dgrad/wrw) vs aten/MIOpen.

Runs every distinct ResNet-50 conv shape at the bench batch size and
prints one row per (shape, op): native ms, aten ms, ratio, TFLOP/s.
Run on a GPU box:  python tools/bench_conv_shapes.py [--batch 256]
"""

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

# (Ci, H, W, Co, k, stride, pad, count) — ResNet-50 distinct conv shapes
R50_SHAPES = [
    (64, 56, 56, 64, 1, 1, 0, 1),
    (64, 56, 56, 64, 3, 1, 1, 3),
    (64, 56, 56, 256, 1, 1, 0, 4),
    (256, 56, 56, 64, 1, 1, 0, 2),
    (256, 56, 56, 128, 1, 1, 0, 1),
    (128, 56, 56, 128, 3, 2, 1, 1),
    (256, 56, 56, 512, 1, 2, 0, 1),
    (128, 28, 28, 512, 1, 1, 0, 4),
    (512, 28, 28, 128, 1, 1, 0, 3),
    (128, 28, 28, 128, 3, 1, 1, 3),
    (512, 28, 28, 256, 1, 1, 0, 1),
    (256, 28, 28, 256, 3, 2, 1, 1),
    (512, 28, 28, 1024, 1, 2, 0, 1),
    (256, 14, 14, 1024, 1, 1, 0, 6),
    (1024, 14, 14, 256, 1, 1, 0, 5),
    (256, 14, 14, 256, 3, 1, 1, 5),
    (1024, 14, 14, 512, 1, 1, 0, 1),
    (512, 14, 14, 512, 3, 2, 1, 1),
    (1024, 14, 14, 2048, 1, 2, 0, 1),
    (512, 7, 7, 2048, 1, 1, 0, 3),
    (2048, 7, 7, 512, 1, 1, 0, 2),
    (512, 7, 7, 512, 3, 1, 1, 2),
]


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=256)
    args = ap.parse_args()
    from deeplearning_cfn_amd.ops import native

    n = native()
    b = args.batch
    tot = {"fwd": [0.0, 0.0], "dgrad": [0.0, 0.0], "wrw": [0.0, 0.0]}
    print(f"batch={b}  (ms per call; ratio>1 = native faster)")
    print(f"{'shape':32s} {'op':6s} {'native':>8s} {'aten':>8s} "
          f"{'ratio':>6s} {'nat TF/s':>9s}")
    for ci, h, w, co, k, s, p, count in R50_SHAPES:
        x = torch.randn(b, ci, h, w, device="cuda").bfloat16()
        x = x.contiguous(memory_format=torch.channels_last)
        wt = (torch.randn(co, ci, k, k, device="cuda") * 0.05).bfloat16()
        wt = wt.contiguous(memory_format=torch.channels_last)
        ho = (h + 2 * p - k) // s + 1
        dy = torch.randn(b, co, ho, ho, device="cuda").bfloat16()
        dy = dy.contiguous(memory_format=torch.channels_last)
        flops = 2.0 * b * ho * ho * co * ci * k * k
        name = f"{ci}x{h}^2->{co} k{k}s{s}"

        ops = {
            "fwd": (lambda: n.conv2d_fwd(x, wt, s, p),
                    lambda: torch.nn.functional.conv2d(x, wt, None, s, p)),
            "dgrad": (lambda: n.conv2d_dgrad(dy, wt, h, w, s, p),
                      lambda: torch.ops.aten.convolution_backward(
                          dy, x, wt, None, [s, s], [p, p], [1, 1], False,
                          [0, 0], 1, [True, False, False])[0]),
            "wrw": (lambda: n.conv2d_wrw(dy, x, k, k, s, p),
                    lambda: torch.ops.aten.convolution_backward(
                        dy, x, wt, None, [s, s], [p, p], [1, 1], False,
                        [0, 0], 1, [False, True, False])[1]),
        }
        for opname, (fn_nat, fn_aten) in ops.items():
            tn = timeit(fn_nat)
            ta = timeit(fn_aten)
            tot[opname][0] += tn * count
            tot[opname][1] += ta * count
            print(f"{name:32s} {opname:6s} {tn*1e3:8.3f} {ta*1e3:8.3f} "
                  f"{ta/tn:6.2f} {flops/tn/1e12:9.1f}", flush=True)
    print("\nweighted totals over ResNet-50 (ms/step contribution):")
    for opname, (tn, ta) in tot.items():
        print(f"  {opname:6s} native {tn*1e3:8.2f}  aten {ta*1e3:8.2f}  "
              f"ratio {ta/max(tn,1e-12):5.2f}")


if __name__ == "__main__":
    main()
