#!/usr/bin/env python3
"""Interleaved within-process A/B bench (guide §5.4 rule 24).

Cross-process perf deltas < 10% are noise (DVFS drift between runs was
repeatedly observed at 3-8% in round 1 — see profiles/BENCH_HISTORY.md
dead-ends).  This harness runs N variants x M rounds interleaved in ONE
process and reports median/min per variant.

Usage (GPU box):
  python tools/ab_bench.py --op wrw --shapes "256,14,1024,1,1" --rounds 5
  python tools/ab_bench.py --op step --rounds 4   # whole training step
"""

import argparse
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402


def timed(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def make_conv_op(op, ci, hw, co, k, s):
    from deeplearning_cfn_amd.ops import native
    n = native()
    b = 1024
    p = k // 2
    x = torch.randn(b, ci, hw, hw, device="cuda").bfloat16()
    x = x.contiguous(memory_format=torch.channels_last)
    w = (torch.randn(co, ci, k, k, device="cuda") * 0.05).bfloat16()
    w = w.contiguous(memory_format=torch.channels_last)
    ho = (hw + 2 * p - k) // s + 1
    dy = torch.randn(b, co, ho, ho, device="cuda").bfloat16()
    dy = dy.contiguous(memory_format=torch.channels_last)
    if op == "fwd":
        return lambda: n.conv2d_fwd(x, w, s, p)
    if op == "dgrad":
        return lambda: n.conv2d_dgrad(dy, w, hw, hw, s, p)
    if op == "wrw":
        return lambda: n.conv2d_wrw(dy, x, k, k, s, p)
    raise ValueError(op)


def make_step_op():
    from deeplearning_cfn_amd.engine import DataParallel
    from deeplearning_cfn_amd.engine.optim import FlatSGD
    from deeplearning_cfn_amd.models import build_resnet
    from deeplearning_cfn_amd.ops import functional as F_amd
    from deeplearning_cfn_amd.utils import to_compute_dtype

    model = build_resnet(50).cuda().to(memory_format=torch.channels_last)
    model = to_compute_dtype(model, torch.bfloat16)
    engine = DataParallel(model, mode="local")
    opt = FlatSGD(engine, lr=0.1)
    x = torch.randn(256, 3, 224, 224, device="cuda").bfloat16()
    x = x.contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (256,), device="cuda")

    def step():
        engine.zero_grad()
        loss = F_amd.softmax_cross_entropy(engine(x), y)
        loss.backward()
        engine.finalize_backward()
        opt.step()

    return step


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--op", default="wrw",
                    help="fwd|dgrad|wrw|step")
    ap.add_argument("--shapes", default="256,14,1024,1,1",
                    help="ci,hw,co,k,s (conv ops)")
    ap.add_argument("--rounds", type=int, default=5)
    ap.add_argument("--variants", default="A",
                    help="comma list of labels; set env per variant via "
                         "VAR=<label> hooks in your own edit, or use this "
                         "to measure run-to-run noise of one config")
    args = ap.parse_args()

    if args.op == "step":
        fn = make_step_op()
    else:
        ci, hw, co, k, s = map(int, args.shapes.split(","))
        fn = make_conv_op(args.op, ci, hw, co, k, s)

    labels = args.variants.split(",")
    results = {v: [] for v in labels}
    for r in range(args.rounds):
        for v in labels:
            results[v].append(timed(fn) * 1e3)
    for v in labels:
        xs = results[v]
        print(f"{v}: median {statistics.median(xs):.3f} ms  "
              f"min {min(xs):.3f}  all {['%.3f' % x for x in xs]}")


if __name__ == "__main__":
    main()
