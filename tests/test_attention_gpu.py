"""Fused attention forward (attention.hip) numerics — VERDICT r1 #5.

Checklist from the round-1 draft: parity vs fp32 torch SDPA at
S=128/512, asymmetric (transpose-detecting) fills, repeat-determinism,
and backward parity of the recompute-composition autograd Function.
"""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _native():
    from deeplearning_cfn_amd.ops import native
    return native()


def _ref_sdpa_fp32(q, k, v):
    scale = q.shape[-1] ** -0.5
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    p = torch.softmax(s, dim=-1)
    return p @ v.float()


@pytest.mark.parametrize("B,H,S", [(2, 4, 128), (1, 12, 512), (3, 2, 64)])
def test_attn_fwd_parity(B, H, S):
    torch.manual_seed(0)
    q = torch.randn(B, H, S, 64, device="cuda").bfloat16()
    k = torch.randn(B, H, S, 64, device="cuda").bfloat16()
    v = torch.randn(B, H, S, 64, device="cuda").bfloat16()
    o, lse = _native().attn_fwd(q, k, v, 64 ** -0.5)
    ref = _ref_sdpa_fp32(q, k, v)
    err = (o.float() - ref).abs().max().item()
    assert err < 2e-2, err
    # lse parity
    s = (q.float() @ k.float().transpose(-1, -2)) * (64 ** -0.5)
    lse_ref = torch.logsumexp(s, dim=-1)
    assert (lse - lse_ref).abs().max().item() < 1e-2


def test_attn_fwd_asymmetric():
    """Transpose-detecting fill: distinct ramps along every axis."""
    B, H, S = 1, 2, 128
    q = torch.zeros(B, H, S, 64, device="cuda")
    k = torch.zeros_like(q)
    v = torch.zeros_like(q)
    ar_s = torch.arange(S, device="cuda").float()
    ar_d = torch.arange(64, device="cuda").float()
    q += 0.01 * ar_s[:, None] + 0.003 * ar_d[None, :]
    k += 0.007 * ar_s[:, None] - 0.002 * ar_d[None, :]
    v += torch.sin(0.1 * ar_s)[:, None] + 0.05 * ar_d[None, :]
    q[0, 1] *= -0.5
    k[0, 1] += 0.1
    q, k, v = q.bfloat16(), k.bfloat16(), v.bfloat16()
    o, _ = _native().attn_fwd(q, k, v, 64 ** -0.5)
    ref = _ref_sdpa_fp32(q, k, v)
    assert (o.float() - ref).abs().max().item() < 2e-2


def test_attn_fwd_deterministic():
    torch.manual_seed(1)
    q = torch.randn(2, 4, 256, 64, device="cuda").bfloat16()
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    o0, l0 = _native().attn_fwd(q, k, v, 0.125)
    for _ in range(5):
        o, l = _native().attn_fwd(q, k, v, 0.125)
        assert torch.equal(o, o0) and torch.equal(l, l0)


def test_attn_autograd_backward_parity():
    from deeplearning_cfn_amd.ops import functional as F_amd
    torch.manual_seed(2)
    B, H, S = 2, 3, 128
    qkv = [torch.randn(B, H, S, 64, device="cuda").bfloat16()
               .requires_grad_(True) for _ in range(3)]
    o = F_amd.attention(*qkv)
    g = torch.randn_like(o)
    o.backward(g)
    grads = [t.grad.clone() for t in qkv]

    # fp32 composition reference
    refs = [t.detach().float().requires_grad_(True) for t in qkv]
    o_ref = _ref_sdpa_fp32(*[r for r in refs])
    o_ref.backward(g.float())
    for got, ref in zip(grads, refs):
        err = (got.float() - ref.grad).abs().max().item()
        assert err < 5e-2, err


@pytest.mark.parametrize("B,H,S", [(2, 4, 128), (1, 8, 512)])
def test_attn_bwd_kernel_parity(B, H, S):
    """Raw fused backward vs fp32 autograd through the composition."""
    torch.manual_seed(5)
    q = torch.randn(B, H, S, 64, device="cuda").bfloat16()
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    scale = 64 ** -0.5
    o, lse = _native().attn_fwd(q, k, v, scale)
    g = torch.randn_like(o)
    dq, dk, dv = _native().attn_bwd(g, q, k, v, o, lse, scale)

    refs = [t.detach().float().requires_grad_(True) for t in (q, k, v)]
    o_ref = _ref_sdpa_fp32(*refs)
    o_ref.backward(g.float())
    for got, ref in zip((dq, dk, dv), refs):
        err = (got.float() - ref.grad).abs().max().item()
        assert err < 6e-2, err


def test_attn_bwd_deterministic_dkdv():
    """dK/dV are plain stores (deterministic); dQ uses fp32 atomics and
    is documented as order-dependent — only dk/dv are checked bitwise."""
    torch.manual_seed(6)
    q = torch.randn(2, 4, 256, 64, device="cuda").bfloat16()
    k, v = torch.randn_like(q), torch.randn_like(q)
    o, lse = _native().attn_fwd(q, k, v, 0.125)
    g = torch.randn_like(o)
    _, dk0, dv0 = _native().attn_bwd(g, q, k, v, o, lse, 0.125)
    for _ in range(3):
        _, dk, dv = _native().attn_bwd(g, q, k, v, o, lse, 0.125)
        assert torch.equal(dk, dk0) and torch.equal(dv, dv0)


def test_attn_bert_block_uses_fused():
    """BERT SelfAttention routes through the fused kernel on GPU."""
    from deeplearning_cfn_amd.models.bert import BertConfig, SelfAttention
    from deeplearning_cfn_amd.utils import to_compute_dtype
    torch.manual_seed(3)
    cfg = BertConfig(hidden=128, heads=2, layers=1)
    attn = SelfAttention(cfg).cuda()
    attn = to_compute_dtype(attn, torch.bfloat16)
    x = torch.randn(2, 128, 128, device="cuda").bfloat16()
    y = attn(x)
    loss = y.float().square().mean()
    loss.backward()
    assert torch.isfinite(loss).item()
    for p in attn.parameters():
        assert p.grad is not None and torch.isfinite(p.grad.float()).all()
