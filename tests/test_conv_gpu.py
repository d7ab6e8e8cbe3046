"""Implicit-GEMM conv numerics vs fp32 torch conv (GPU-only)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

# (N, Ci, H, W, Co, k, stride, pad) — the ResNet-50 shape classes
SHAPES = [
    (4, 64, 56, 56, 64, 1, 1, 0),       # bottleneck 1x1 reduce
    (4, 64, 56, 56, 256, 1, 1, 0),      # 1x1 expand
    (4, 64, 56, 56, 64, 3, 1, 1),       # 3x3
    (4, 256, 56, 56, 512, 1, 2, 0),     # strided downsample 1x1
    (4, 128, 28, 28, 128, 3, 2, 1),     # strided 3x3
    (2, 2048, 7, 7, 512, 1, 1, 0),      # deep narrow
    (2, 16, 32, 32, 16, 3, 1, 1),       # CIFAR shape
    (2, 3, 224, 224, 64, 7, 2, 3),      # ImageNet stem (small-Cin path)
    (2, 3, 32, 32, 16, 3, 1, 1),        # CIFAR stem
]


def _native():
    from deeplearning_cfn_amd.ops import native
    return native()


def _mk(n, ci, h, w, co, k, seed):
    g = torch.Generator(device="cuda").manual_seed(seed)
    x = torch.randn(n, ci, h, w, generator=g, device="cuda").bfloat16()
    x = x.contiguous(memory_format=torch.channels_last)
    wt = (torch.randn(co, ci, k, k, generator=g, device="cuda") *
          (ci * k * k) ** -0.5).bfloat16()
    return x, wt


@pytest.mark.parametrize("shape", SHAPES)
def test_conv_fwd_parity(shape):
    n, ci, h, w, co, k, s, p = shape
    x, wt = _mk(n, ci, h, w, co, k, seed=sum(shape))
    y = _native().conv2d_fwd(x, wt, s, p)
    ref = torch.nn.functional.conv2d(x.float(), wt.float(), None, s, p)
    assert y.is_contiguous(memory_format=torch.channels_last)
    torch.testing.assert_close(y.float(), ref, rtol=3e-2,
                               atol=3e-2 * (ci * k * k) ** 0.25)


@pytest.mark.parametrize("shape", SHAPES)
def test_conv_dgrad_parity(shape):
    n, ci, h, w, co, k, s, p = shape
    x, wt = _mk(n, ci, h, w, co, k, seed=sum(shape) + 1)
    xf = x.float().requires_grad_()
    ref = torch.nn.functional.conv2d(xf, wt.float(), None, s, p)
    g = torch.Generator(device="cuda").manual_seed(7)
    dy = torch.randn(ref.shape, generator=g, device="cuda").bfloat16()
    dy = dy.contiguous(memory_format=torch.channels_last)
    ref.backward(dy.float())
    dx = _native().conv2d_dgrad(dy, wt, h, w, s, p)
    torch.testing.assert_close(dx.float(), xf.grad, rtol=3e-2,
                               atol=3e-2 * (co * k * k) ** 0.25)


@pytest.mark.parametrize("shape", SHAPES)
def test_conv_wrw_parity(shape):
    n, ci, h, w, co, k, s, p = shape
    x, wt = _mk(n, ci, h, w, co, k, seed=sum(shape) + 2)
    wf = wt.float().requires_grad_()
    ref = torch.nn.functional.conv2d(x.float(), wf, None, s, p)
    g = torch.Generator(device="cuda").manual_seed(11)
    dy = torch.randn(ref.shape, generator=g, device="cuda").bfloat16()
    dy = dy.contiguous(memory_format=torch.channels_last)
    ref.backward(dy.float())
    dw = _native().conv2d_wrw(dy, x, k, k, s, p)
    m = n * ref.shape[2] * ref.shape[3]
    torch.testing.assert_close(dw.float(), wf.grad, rtol=4e-2,
                               atol=2e-2 * m ** 0.5)


def test_conv_autograd_function():
    from deeplearning_cfn_amd.ops import functional as F_amd
    n, ci, h, w, co, k, s, p = 2, 64, 14, 14, 128, 3, 1, 1
    x, wt = _mk(n, ci, h, w, co, k, seed=42)
    x = x.requires_grad_()
    wt = wt.requires_grad_()
    y = F_amd.conv2d(x, wt, s, p)
    xf = x.detach().float().requires_grad_()
    wf = wt.detach().float().requires_grad_()
    ref = torch.nn.functional.conv2d(xf, wf, None, s, p)
    torch.testing.assert_close(y.float(), ref, rtol=3e-2, atol=0.5)
    gg = torch.Generator(device="cuda").manual_seed(3)
    dy = torch.randn(ref.shape, generator=gg, device="cuda").bfloat16()
    y.backward(dy)
    ref.backward(dy.float())
    torch.testing.assert_close(x.grad.float(), xf.grad, rtol=3e-2, atol=0.6)
    torch.testing.assert_close(wt.grad.float(), wf.grad, rtol=3e-2,
                               atol=3.0)  # wgrad sums over N*H*W rows
