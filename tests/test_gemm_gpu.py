"""MFMA GEMM numerics vs fp32 torch matmul (GPU-only).

Transpose-detecting inputs per guide §5.4 rule 16: identity-A with an
ASYMMETRIC B, plus full random parity at several shapes including edge
(non-multiple-of-tile) cases.
"""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _native():
    from deeplearning_cfn_amd.ops import native
    return native()


def test_gemm_identity_asymmetric():
    n = _native()
    K = 128
    A = torch.eye(K, device="cuda", dtype=torch.bfloat16)
    # B[N,K] asymmetric: B[i][j] = i*3 + j*0.125
    i = torch.arange(128, device="cuda").unsqueeze(1)
    j = torch.arange(K, device="cuda").unsqueeze(0)
    B = (i * 3 + j * 0.125).bfloat16()
    C = n.gemm_nt(A, B)  # I @ B^T = B^T
    torch.testing.assert_close(C.float(), B.float().t(), rtol=0, atol=0)


@pytest.mark.parametrize("M,N,K", [
    (128, 128, 64),       # single tile
    (256, 384, 512),      # interior multi-tile
    (256, 1000, 2048),    # the ResNet-50 FC shape (N edge)
    (100, 130, 72),       # all-edge
    (512, 768, 3072),     # BERT MLP shape
])
def test_gemm_random_parity(M, N, K):
    n = _native()
    g = torch.Generator(device="cuda").manual_seed(M * 7 + N)
    A = torch.randn(M, K, generator=g, device="cuda").bfloat16()
    B = torch.randn(N, K, generator=g, device="cuda").bfloat16()
    C = n.gemm_nt(A, B)
    ref = A.float() @ B.float().t()
    # bf16 inputs, fp32 accumulate: error ~ sqrt(K) * eps_bf16 * |a||b|
    torch.testing.assert_close(C.float(), ref, rtol=2e-2,
                               atol=2e-2 * (K ** 0.5))


def test_gemm_nn_tn():
    n = _native()
    g = torch.Generator(device="cuda").manual_seed(99)
    G = torch.randn(64, 96, generator=g, device="cuda").bfloat16()
    W = torch.randn(96, 160, generator=g, device="cuda").bfloat16()
    X = torch.randn(64, 160, generator=g, device="cuda").bfloat16()
    dx = n.gemm_nn(G, W)
    torch.testing.assert_close(dx.float(), G.float() @ W.float(),
                               rtol=2e-2, atol=0.3)
    dw = n.gemm_tn(G, X)
    torch.testing.assert_close(dw.float(), G.float().t() @ X.float(),
                               rtol=2e-2, atol=0.3)


def test_linear_autograd_end_to_end():
    from deeplearning_cfn_amd.ops import functional as F_amd
    g = torch.Generator(device="cuda").manual_seed(5)
    x = torch.randn(256, 512, generator=g, device="cuda").bfloat16().requires_grad_()
    w = torch.randn(384, 512, generator=g, device="cuda").bfloat16().requires_grad_()
    b = torch.randn(384, generator=g, device="cuda").bfloat16().requires_grad_()
    y = F_amd.linear(x, w, b)
    xf = x.detach().float().requires_grad_()
    wf = w.detach().float().requires_grad_()
    bf = b.detach().float().requires_grad_()
    yref = torch.nn.functional.linear(xf, wf, bf)
    torch.testing.assert_close(y.float(), yref, rtol=3e-2, atol=0.5)
    dy = torch.randn(256, 384, generator=g, device="cuda").bfloat16()
    y.backward(dy)
    yref.backward(dy.float())
    torch.testing.assert_close(x.grad.float(), xf.grad, rtol=3e-2, atol=0.6)
    torch.testing.assert_close(w.grad.float(), wf.grad, rtol=3e-2, atol=0.8)
    torch.testing.assert_close(b.grad.float(), bf.grad, rtol=3e-2, atol=0.5)
