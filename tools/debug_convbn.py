#!/usr/bin/env python3
"""Quantify fused-vs-unfused and fused-vs-fp32 grad differences."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch


def stats(name, a, b):
    d = (a.float() - b.float()).abs()
    bb = b.float().abs()
    rel = d.norm() / (b.float().norm() + 1e-12)
    band = d > (2e-2 + 2e-2 * bb)
    q = torch.quantile(d.flatten().float(), torch.tensor(
        [0.5, 0.99, 0.999, 1.0], device=d.device))
    print(f"  {name:10s} relnorm={rel.item():.5f} frac>band="
          f"{band.float().mean().item():.5f} q50={q[0]:.2e} "
          f"q99={q[1]:.2e} q999={q[2]:.2e} max={q[3]:.2e} "
          f"|b|max={bb.max().item():.2e}")


def run(mode):
    from deeplearning_cfn_amd.ops import functional as F_amd
    torch.manual_seed(11)
    N, Ci, H, W, Co = 4, 64, 14, 14, 128
    x = torch.randn(N, Ci, H, W, device="cuda").bfloat16()
    x = x.contiguous(memory_format=torch.channels_last).requires_grad_()
    wt = (torch.randn(Co, Ci, 3, 3, device="cuda") * (Ci * 9) ** -0.5)
    wt = wt.bfloat16().contiguous(memory_format=torch.channels_last)
    wt.requires_grad_()
    bn = torch.nn.BatchNorm2d(Co).cuda()
    bn2 = torch.nn.BatchNorm2d(Co).cuda()
    with torch.no_grad():
        bn.weight.uniform_(0.5, 1.5)
        bn.bias.uniform_(-0.5, 0.5)
        bn2.weight.copy_(bn.weight)
        bn2.bias.copy_(bn.bias)
    res = res2 = None
    if mode == "res":
        res = torch.randn(N, Co, H, W, device="cuda").bfloat16()
        res = res.contiguous(memory_format=torch.channels_last)
        res.requires_grad_()
        res2 = res.detach().clone().requires_grad_()
    y = F_amd.conv_bn_act(x, wt, bn, 1, 1, relu=(mode != "plain"),
                          residual=res, training=True)
    x2 = x.detach().clone().requires_grad_()
    w2 = wt.detach().clone().requires_grad_()
    out = F_amd.conv2d(x2, w2, 1, 1)
    if mode == "res":
        ref = F_amd.batch_norm_add_relu(out, bn2, res2, training=True)
    else:
        ref = F_amd.batch_norm_act(out, bn2, relu=(mode != "plain"),
                                   training=True)
    print(f"mode={mode}")
    stats("y", y, ref)
    stats("rmean", bn.running_mean, bn2.running_mean)
    stats("rvar", bn.running_var, bn2.running_var)
    g = torch.randn_like(ref)
    y.backward(g)
    ref.backward(g)
    stats("x.grad", x.grad, x2.grad)
    stats("w.grad", wt.grad, w2.grad)
    stats("gamma.g", bn.weight.grad, bn2.weight.grad)
    stats("beta.g", bn.bias.grad, bn2.bias.grad)
    if mode == "res":
        stats("res.grad", res.grad, res2.grad)


if __name__ == "__main__":
    for m in ["plain", "relu", "res"]:
        run(m)
