#!/usr/bin/env python3
"""Localize the bn mask-vs-y backward mismatch (GPU)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch


def run(C, shape_hw, dtype, seed=5):
    from deeplearning_cfn_amd.ops import native
    n = native()
    N, H, W = 4, shape_hw, shape_hw
    g = torch.Generator(device="cuda").manual_seed(seed)
    xf = torch.randn(N, C, H, W, generator=g, device="cuda", dtype=torch.float32)
    xf = xf.to(dtype).contiguous(memory_format=torch.channels_last)
    w = (torch.randn(C, generator=g, device="cuda") * 0.5 + 1.0)
    b = torch.randn(C, generator=g, device="cuda") * 0.1
    rm, rv = torch.zeros(C, device="cuda"), torch.ones(C, device="cuda")
    y, mean, invstd, mask = n.bn_fwd_train(xf, w, b, rm, rv, 0.1, 1e-5, True)
    dy = torch.randn(N, C, H, W, generator=g, device="cuda",
                     dtype=torch.float32).to(dtype)
    dy = dy.contiguous(memory_format=torch.channels_last)

    vw = 8 if dtype == torch.bfloat16 else 4
    R = N * H * W
    CU = C // vw
    # expand mask bits -> [R, C]
    mb = mask.view(R, CU).unsqueeze(-1)
    bits = (mb >> torch.arange(vw, device="cuda", dtype=torch.uint8)) & 1
    mask_full = bits.reshape(R, C).bool()
    # y>0 in channels-last order
    ypos = (y.permute(0, 2, 3, 1).reshape(R, C) > 0)
    nm = (mask_full != ypos).sum().item()
    print(f"C={C} {dtype}: mask-vs-(y>0) mismatches: {nm} / {R*C}")
    if nm:
        idx = (mask_full != ypos).nonzero()[:5]
        for r, c in idx.tolist():
            print(f"   r={r} c={c} y={y.permute(0,2,3,1).reshape(R,C)[r,c].item()} "
                  f"mask={mask_full[r,c].item()}")

    dx, dw, db = n.bn_bwd(dy, xf, w, mean, invstd, y, True)
    dx2, dw2, db2 = n.bn_bwd_m(dy, xf, w, mean, invstd, mask)
    d = (dx2.float() - dx.float()).abs()
    print(f"   dx max diff {d.max().item():.3e}  dw {(dw2-dw).abs().max().item():.3e} "
          f"db {(db2-db).abs().max().item():.3e}")


if __name__ == "__main__":
    run(32, 6, torch.float32)
    run(64, 8, torch.bfloat16)
    run(256, 14, torch.bfloat16)
