"""Flat fused optimizers vs torch.optim references (CPU fallback path)."""

import torch

from deeplearning_cfn_amd.engine.ddp import DataParallel
from deeplearning_cfn_amd.engine.optim import FlatAdam, FlatSGD


def _pair(seed=11):
    torch.manual_seed(seed)
    a = torch.nn.Sequential(torch.nn.Linear(16, 64), torch.nn.Tanh(),
                            torch.nn.Linear(64, 8))
    b = torch.nn.Sequential(torch.nn.Linear(16, 64), torch.nn.Tanh(),
                            torch.nn.Linear(64, 8))
    b.load_state_dict(a.state_dict())
    return a, b


def test_flat_sgd_matches_torch():
    a, b = _pair()
    eng = DataParallel(a, mode="local")
    opt = FlatSGD(eng, lr=0.05, momentum=0.9, weight_decay=1e-3)
    ref = torch.optim.SGD(b.parameters(), lr=0.05, momentum=0.9,
                          weight_decay=1e-3)
    for step in range(5):
        x = torch.randn(8, 16, generator=torch.Generator().manual_seed(step))
        eng.zero_grad()
        eng(x).pow(2).mean().backward()
        eng.finalize_backward()
        opt.step()
        ref.zero_grad()
        b(x).pow(2).mean().backward()
        ref.step()
    for p, q in zip(a.parameters(), b.parameters()):
        torch.testing.assert_close(p, q, rtol=1e-5, atol=1e-6)


def test_flat_sgd_nesterov():
    a, b = _pair(12)
    eng = DataParallel(a, mode="local")
    opt = FlatSGD(eng, lr=0.01, momentum=0.9, weight_decay=0.0,
                  nesterov=True)
    ref = torch.optim.SGD(b.parameters(), lr=0.01, momentum=0.9,
                          nesterov=True)
    for step in range(3):
        x = torch.randn(4, 16, generator=torch.Generator().manual_seed(step))
        eng.zero_grad()
        eng(x).sum().backward()
        eng.finalize_backward()
        opt.step()
        ref.zero_grad()
        b(x).sum().backward()
        ref.step()
    for p, q in zip(a.parameters(), b.parameters()):
        torch.testing.assert_close(p, q, rtol=1e-5, atol=1e-6)


def test_flat_adamw_matches_torch():
    a, b = _pair(13)
    eng = DataParallel(a, mode="local")
    opt = FlatAdam(eng, lr=1e-3, weight_decay=0.01)
    ref = torch.optim.AdamW(b.parameters(), lr=1e-3, weight_decay=0.01,
                            eps=1e-8)
    for step in range(5):
        x = torch.randn(8, 16, generator=torch.Generator().manual_seed(step))
        eng.zero_grad()
        eng(x).pow(2).mean().backward()
        eng.finalize_backward()
        opt.step()
        ref.zero_grad()
        b(x).pow(2).mean().backward()
        ref.step()
    for p, q in zip(a.parameters(), b.parameters()):
        torch.testing.assert_close(p, q, rtol=1e-4, atol=1e-5)


def test_state_roundtrip():
    a, _ = _pair(14)
    eng = DataParallel(a, mode="local")
    opt = FlatSGD(eng, lr=0.1, momentum=0.9)
    eng.zero_grad()
    eng(torch.randn(2, 16)).sum().backward()
    eng.finalize_backward()
    opt.step()
    sd = opt.state_dict()
    a2, _ = _pair(14)
    eng2 = DataParallel(a2, mode="local")
    opt2 = FlatSGD(eng2, lr=0.2, momentum=0.5)
    opt2.load_state_dict(sd)
    assert opt2.lr == 0.1 and opt2.momentum == 0.9
    for x, y in zip(opt.momentum_bufs, opt2.momentum_bufs):
        torch.testing.assert_close(x, y)
