#!/usr/bin/env python3
"""Per-op microbenchmarks on the MI355X: native HIP vs torch reference.

Prints one line per op: time, achieved GB/s or TFLOP/s, and the torch
comparison.  Run on a GPU box:  python tools/microbench.py [--csv out]
"""

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def row(name, t_native, t_torch, bytes_=None, flops=None):
    cols = [f"{name:36s}", f"native {t_native*1e3:8.3f} ms"]
    if t_torch is not None:
        cols.append(f"torch {t_torch*1e3:8.3f} ms  ({t_torch/t_native:4.2f}x)")
    if bytes_:
        cols.append(f"{bytes_/t_native/1e12:6.2f} TB/s")
    if flops:
        cols.append(f"{flops/t_native/1e12:7.1f} TF/s")
    print("  ".join(cols), flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()
    from deeplearning_cfn_amd.ops import native
    n = native()
    g = torch.Generator(device="cuda").manual_seed(0)

    def rand(shape, dtype=torch.bfloat16, cl=False):
        x = torch.randn(*shape, generator=g, device="cuda").to(dtype)
        return x.contiguous(memory_format=torch.channels_last) if cl else x

    # --- elementwise ---
    for shape in [(256, 256, 56, 56), (256, 2048, 7, 7)]:
        a, b = rand(shape, cl=True), rand(shape, cl=True)
        nb = a.numel() * 2 * 3  # r+r+w bf16
        t1 = timeit(lambda: n.add_relu_fwd(a, b), args.iters)
        t2 = timeit(lambda: torch.relu(a + b), args.iters)
        row(f"add_relu {shape}", t1, t2, bytes_=nb)

    # --- batchnorm fwd/bwd, ResNet-50 hot shapes ---
    for shape in [(256, 64, 56, 56), (256, 256, 56, 56), (256, 512, 28, 28),
                  (256, 1024, 14, 14), (256, 2048, 7, 7)]:
        C = shape[1]
        x = rand(shape, cl=True)
        w = torch.rand(C, device="cuda") + 0.5
        bb = torch.randn(C, device="cuda") * 0.1
        rm, rv = torch.zeros(C, device="cuda"), torch.ones(C, device="cuda")
        nb = x.numel() * 2 * 3  # ~2 reads + 1 write
        t1 = timeit(lambda: n.bn_fwd_train(x, w, bb, rm, rv, 0.1, 1e-5, True),
                    args.iters)
        xf = x.float()
        t2 = timeit(lambda: torch.relu(torch.nn.functional.batch_norm(
            x, rm, rv, w, bb, True, 0.1, 1e-5)), args.iters)
        row(f"bn_relu_fwd {shape}", t1, t2, bytes_=nb)
        y, mean, invstd = n.bn_fwd_train(x, w, bb, rm, rv, 0.1, 1e-5, True)
        dy = rand(shape, cl=True)
        t1 = timeit(lambda: n.bn_bwd(dy, x, w, mean, invstd, y, True),
                    args.iters)
        row(f"bn_relu_bwd {shape}", t1, None, bytes_=x.numel() * 2 * 5)

    # --- softmax CE ---
    logits = rand((256, 1000))
    tgt = torch.randint(0, 1000, (256,), device="cuda")
    t1 = timeit(lambda: n.softmax_ce_fwd(logits, tgt), args.iters)
    t2 = timeit(lambda: torch.nn.functional.cross_entropy(logits.float(), tgt),
                args.iters)
    row("softmax_ce_fwd (256,1000)", t1, t2)

    # --- fused SGD over a ResNet-50-sized flat bucket ---
    N = 25_600_000
    p = torch.randn(N, device="cuda").bfloat16()
    gr = torch.randn(N, device="cuda").bfloat16()
    m = torch.zeros(N, device="cuda")
    master = torch.randn(N, device="cuda")
    t1 = timeit(lambda: n.sgd_momentum_flat_master(p, gr, m, master, 0.1,
                                                   0.9, 1e-4, False),
                args.iters)
    row("sgd_flat_master 25.6M", t1, None,
        bytes_=N * (2 + 2 + 4 * 2 + 4 * 2))

    # --- MFMA GEMM ---
    for M, N_, K in [(4096, 4096, 4096), (8192, 8192, 8192),
                     (256, 1000, 2048), (512, 3072, 768)]:
        A = rand((M, K))
        B = rand((N_, K))
        fl = 2.0 * M * N_ * K
        t1 = timeit(lambda: n.gemm_nt(A, B), args.iters)
        t2 = timeit(lambda: A @ B.t(), args.iters)
        row(f"gemm_nt {M}x{N_}x{K}", t1, t2, flops=fl)


if __name__ == "__main__":
    main()
