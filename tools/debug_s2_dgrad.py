#!/usr/bin/env python3
"""Localize the stride-2 dgrad parity-decomposition failure (GPU)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch


def check(n, ci, h, w, co, k, s, p, verbose=False):
    from deeplearning_cfn_amd.ops import native
    g = torch.Generator(device="cuda").manual_seed(7)
    x = torch.randn(n, ci, h, w, generator=g, device="cuda").bfloat16()
    x = x.contiguous(memory_format=torch.channels_last)
    wt = (torch.randn(co, ci, k, k, generator=g, device="cuda") *
          (ci * k * k) ** -0.5).bfloat16()
    xf = x.float().requires_grad_()
    ref = torch.nn.functional.conv2d(xf, wt.float(), None, s, p)
    dy = torch.randn(ref.shape, generator=g, device="cuda").bfloat16()
    dy = dy.contiguous(memory_format=torch.channels_last)
    ref.backward(dy.float())
    dx = native().conv2d_dgrad(dy, wt, h, w, s, p)
    err = (dx.float() - xf.grad).abs()
    print(f"shape ({n},{ci},{h},{w})->{co} k{k}s{s}p{p}: max_err="
          f"{err.max().item():.4f} mean={err.mean().item():.5f} "
          f"cl={dx.is_contiguous(memory_format=torch.channels_last)}")
    if err.max().item() > 0.1:
        # which pixel parities are wrong?
        for a in range(2):
            for b in range(2):
                e = err[:, :, a::2, b::2]
                print(f"  class ({a},{b}): max={e.max().item():.4f} "
                      f"frac_bad={(e > 0.05).float().mean().item():.3f}")
        if verbose:
            bad = (err > 0.05).nonzero()[:10]
            print("  first bad idx:", bad.tolist())
            n0, c0, h0, w0 = bad[0].tolist()
            print("  got", dx.float()[n0, c0, h0, w0].item(),
                  "want", xf.grad[n0, c0, h0, w0].item())


if __name__ == "__main__":
    check(1, 8, 8, 8, 8, 1, 2, 0, verbose=True)
    check(1, 8, 8, 8, 8, 3, 2, 1, verbose=True)
    check(2, 64, 16, 16, 64, 1, 2, 0)
    check(4, 256, 56, 56, 512, 1, 2, 0, verbose=True)
    check(4, 128, 56, 56, 128, 3, 2, 1)
