#!/usr/bin/env python3
"""Race screen for the 8-phase igemm pipeline (guide §5 two-lane rule:
a new sync structure needs a multi-run screen at several sizes, not one
refcheck).  10 rounds x {256, 512, 1024, 4096} GEMM + 2 conv shapes,
fresh random data each round, bitwise-compared across repeats and
checked against fp32 torch."""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    from deeplearning_cfn_amd.ops import native
    n = native()
    g = torch.Generator(device="cuda").manual_seed(0)
    fails = 0
    for rnd in range(10):
        for size in (256, 512, 1024, 4096):
            A = torch.randn(size, size, generator=g, device="cuda").bfloat16()
            B = torch.randn(size, size, generator=g, device="cuda").bfloat16()
            C1 = n.gemm_nt(A, B)
            C2 = n.gemm_nt(A, B)
            if not torch.equal(C1, C2):
                print(f"FAIL nondeterminism gemm {size} round {rnd}")
                fails += 1
            ref = (A.float() @ B.float().t()).bfloat16()
            err = (C1.float() - ref.float()).abs().max().item()
            lim = 0.15 * (size ** 0.5)
            if err > lim:
                print(f"FAIL accuracy gemm {size} round {rnd}: {err:.3f} > {lim:.3f}")
                fails += 1
        # conv shapes through the 256^2 path
        for (nb, ci, h, w, co, k, s_, p) in [(8, 64, 56, 56, 256, 1, 1, 0),
                                             (8, 128, 28, 28, 128, 3, 1, 1)]:
            x = torch.randn(nb, ci, h, w, generator=g, device="cuda").bfloat16()
            x = x.contiguous(memory_format=torch.channels_last)
            wt = (torch.randn(co, ci, k, k, generator=g, device="cuda")
                  * (ci * k * k) ** -0.5).bfloat16()
            y1 = n.conv2d_fwd(x, wt, s_, p)
            y2 = n.conv2d_fwd(x, wt, s_, p)
            if not torch.equal(y1, y2):
                print(f"FAIL nondeterminism conv {ci}x{k} round {rnd}")
                fails += 1
            ref = torch.nn.functional.conv2d(x.float(), wt.float(), None, s_, p)
            err = (y1.float() - ref).abs().max().item()
            if err > 0.2:
                print(f"FAIL accuracy conv {ci}x{k} round {rnd}: {err:.3f}")
                fails += 1
    print("race screen:", "PASS" if fails == 0 else f"{fails} FAILURES")

    # perf of the 8ph kernel
    for size in (4096, 8192):
        A = torch.randn(size, size, generator=g, device="cuda").bfloat16()
        B = torch.randn(size, size, generator=g, device="cuda").bfloat16()
        for _ in range(3):
            n.gemm_nt(A, B)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            n.gemm_nt(A, B)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 10
        print(f"gemm_nt {size}^3: {dt*1e3:.3f} ms = {2*size**3/dt/1e12:.0f} TF/s")
    return 1 if fails else 0


if __name__ == "__main__":
    sys.exit(main())
