#!/usr/bin/env python3
"""Diagnose why MIOpen picks naive_conv kernels under the native-op path.

Hooks every Conv2d in ResNet-50 (bf16, channels_last), records the exact
input / weight / grad_output layouts per impl, then re-times each unique
conv signature standalone.  Run on a GPU box."""

import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def sig(t):
    if t is None:
        return None
    return (tuple(t.shape), tuple(t.stride()), str(t.dtype),
            t.is_contiguous(memory_format=torch.channels_last),
            t.is_contiguous())


def run(impl):
    os.environ["DLX_IMPL"] = impl
    from deeplearning_cfn_amd.engine import DataParallel
    from deeplearning_cfn_amd.engine.optim import FlatSGD
    from deeplearning_cfn_amd.models import build_resnet
    from deeplearning_cfn_amd.ops import functional as F_amd
    from deeplearning_cfn_amd.utils import to_compute_dtype

    torch.manual_seed(0)
    model = build_resnet(50).to("cuda").to(memory_format=torch.channels_last)
    model = to_compute_dtype(model, torch.bfloat16)
    engine = DataParallel(model, mode="local")
    opt = FlatSGD(engine, lr=0.1)

    records = {}

    def pre_hook(name):
        def fn(mod, inputs):
            records.setdefault(name, {})["in"] = sig(inputs[0])
            records[name]["w"] = sig(mod.weight)
        return fn

    def bwd_hook(name):
        def fn(mod, gin, gout):
            records.setdefault(name, {})["gout"] = sig(gout[0])
        return fn

    for name, mod in model.named_modules():
        if isinstance(mod, torch.nn.Conv2d):
            mod.register_forward_pre_hook(pre_hook(name))
            mod.register_full_backward_hook(bwd_hook(name))

    x = torch.randn(32, 3, 224, 224, device="cuda").bfloat16()
    x = x.contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (32,), device="cuda")
    engine.zero_grad()
    loss = F_amd.softmax_cross_entropy(engine(x), y)
    loss.backward()
    engine.finalize_backward()
    opt.step()
    torch.cuda.synchronize()
    os.environ.pop("DLX_IMPL", None)
    return records


def main():
    rec_n = run("auto")
    print("== layouts that are NOT packed channels_last (native impl) ==")
    bad = 0
    for name, r in rec_n.items():
        for key in ("in", "w", "gout"):
            s = r.get(key)
            if s and not s[3] and not s[4]:
                print(f"  {name} {key}: shape={s[0]} stride={s[1]} CL={s[3]} C={s[4]}")
                bad += 1
    print(f"  ({bad} non-packed tensors)")
    # weight check: CL vs NCHW
    wcl = sum(1 for r in rec_n.values() if r["w"][3])
    wc = sum(1 for r in rec_n.values() if r["w"][4])
    print(f"  weights: {wcl}/{len(rec_n)} channels_last, {wc} NCHW-contig")
    incl = sum(1 for r in rec_n.values() if r['in'][3])
    gcl = sum(1 for r in rec_n.values() if r.get('gout') and r['gout'][3])
    print(f"  inputs channels_last: {incl}/{len(rec_n)}; gout CL: {gcl}")

    # standalone timing of a suspect conv with both weight layouts
    xcl = torch.randn(32, 256, 56, 56, device="cuda").bfloat16().contiguous(
        memory_format=torch.channels_last)
    w_nchw = torch.randn(64, 256, 1, 1, device="cuda").bfloat16()
    w_cl = w_nchw.contiguous(memory_format=torch.channels_last)
    for tag, w in [("w NCHW", w_nchw), ("w CL", w_cl)]:
        for _ in range(3):
            torch.nn.functional.conv2d(xcl, w)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            torch.nn.functional.conv2d(xcl, w)
        torch.cuda.synchronize()
        print(f"  conv2d 1x1 256->64 {tag}: {(time.perf_counter()-t0)/10*1e3:.3f} ms")


if __name__ == "__main__":
    main()
