"""Second kernel batch: pooling, pack, LayerNorm, GELU parity (GPU)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _native():
    from deeplearning_cfn_amd.ops import native
    return native()


def _rand(shape, dtype=torch.bfloat16, cl=False, seed=0):
    g = torch.Generator(device="cuda").manual_seed(seed)
    x = torch.randn(*shape, generator=g, device="cuda", dtype=torch.float32)
    x = x.to(dtype)
    if cl:
        x = x.contiguous(memory_format=torch.channels_last)
    return x


@pytest.mark.parametrize("shape", [(4, 64, 112, 112), (2, 16, 15, 17)])
def test_maxpool_fwd_bwd(shape):
    n = _native()
    x = _rand(shape, torch.float32, cl=True, seed=3).requires_grad_()
    y, argmax = n.maxpool3x3s2_fwd(x.detach())
    ref = torch.nn.functional.max_pool2d(x, 3, stride=2, padding=1)
    torch.testing.assert_close(y.float(), ref, rtol=0, atol=0)
    dy = _rand(tuple(ref.shape), torch.float32, cl=True, seed=4)
    ref.backward(dy)
    dx = n.maxpool3x3s2_bwd(dy, argmax, shape[2], shape[3])
    torch.testing.assert_close(dx.float(), x.grad, rtol=1e-5, atol=1e-5)


def test_global_avg_pool():
    n = _native()
    x = _rand((8, 2048, 7, 7), torch.bfloat16, cl=True, seed=5)
    y = n.gap_fwd(x)
    ref = torch.nn.functional.adaptive_avg_pool2d(x.float(), 1)
    torch.testing.assert_close(y.float(), ref, rtol=2e-2, atol=2e-2)
    dy = _rand((8, 2048, 1, 1), torch.bfloat16, seed=6)
    dx = n.gap_bwd(dy, 7, 7)
    ref_dx = dy.float().expand(8, 2048, 7, 7) / 49
    torch.testing.assert_close(dx.float(), ref_dx.contiguous(
        memory_format=torch.channels_last), rtol=2e-2, atol=1e-3)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_pack_roundtrip(dtype):
    n = _native()
    x = _rand((4, 35, 19, 23), dtype)  # odd shapes
    nhwc = n.pack_nchw_to_nhwc(x)
    assert nhwc.is_contiguous(memory_format=torch.channels_last)
    torch.testing.assert_close(nhwc, x, rtol=0, atol=0)  # same logical values
    back = n.pack_nhwc_to_nchw(nhwc)
    assert back.is_contiguous()
    torch.testing.assert_close(back, x, rtol=0, atol=0)


def test_transpose2d():
    n = _native()
    x = _rand((1000, 2048), torch.bfloat16, seed=9)
    y = n.transpose2d(x)
    torch.testing.assert_close(y, x.t().contiguous(), rtol=0, atol=0)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_layernorm_fwd_bwd(dtype):
    n = _native()
    R, D = 512, 768
    x = _rand((R, D), dtype, seed=11)
    w = torch.randn(D, device="cuda") * 0.5 + 1.0
    b = torch.randn(D, device="cuda") * 0.1
    y, mean, invstd = n.ln_fwd(x, w, b, 1e-5)
    xf = x.float().requires_grad_()
    wf = w.clone().requires_grad_()
    bf = b.clone().requires_grad_()
    ref = torch.nn.functional.layer_norm(xf, (D,), wf, bf, 1e-5)
    tol = 3e-2 if dtype == torch.bfloat16 else 1e-4
    torch.testing.assert_close(y.float(), ref, rtol=tol, atol=tol)
    dy = _rand((R, D), dtype, seed=12)
    ref.backward(dy.float())
    dx, dg, db_ = n.ln_bwd(dy, x, w, mean, invstd)
    torch.testing.assert_close(dx.float(), xf.grad, rtol=tol, atol=tol * 2)
    torch.testing.assert_close(dg, wf.grad, rtol=1e-2, atol=0.5 if dtype == torch.bfloat16 else 1e-2)
    torch.testing.assert_close(db_, bf.grad, rtol=1e-2, atol=0.5 if dtype == torch.bfloat16 else 1e-2)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
def test_gelu_fwd_bwd(dtype):
    n = _native()
    x = _rand((1000,), dtype, seed=13)
    y = n.gelu_fwd(x)
    ref = torch.nn.functional.gelu(x.float())
    tol = 2e-2 if dtype == torch.bfloat16 else 1e-5
    torch.testing.assert_close(y.float(), ref, rtol=tol, atol=tol)
    xf = x.float().requires_grad_()
    torch.nn.functional.gelu(xf).sum().backward()
    g = torch.ones_like(x)
    dx = n.gelu_bwd(g, x)
    torch.testing.assert_close(dx.float(), xf.grad, rtol=tol, atol=tol)
