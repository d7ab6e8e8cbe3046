"""End-to-end GPU integration: native model step, impl cross-check,
smoke() contract."""

import json
import os
import subprocess
import sys
from pathlib import Path

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = Path(__file__).resolve().parent.parent


def test_native_extension_loads():
    from deeplearning_cfn_amd.ops import native_available, native
    assert native_available(), "HIP extension must load on a GPU box"
    native()


def _one_step(impl: str, seed=7):
    os.environ["DLX_IMPL"] = impl
    # re-evaluate dispatch each call (impl_mode() reads env lazily)
    import torch
    from deeplearning_cfn_amd.engine import DataParallel
    from deeplearning_cfn_amd.engine.optim import FlatSGD
    from deeplearning_cfn_amd.models import build_resnet
    from deeplearning_cfn_amd.ops import functional as F_amd
    from deeplearning_cfn_amd.utils import to_compute_dtype

    torch.manual_seed(seed)
    model = build_resnet(20, dataset="cifar10").to("cuda")
    model = model.to(memory_format=torch.channels_last)
    model = to_compute_dtype(model, torch.bfloat16)
    engine = DataParallel(model, mode="local")
    opt = FlatSGD(engine, lr=0.05, momentum=0.9)
    g = torch.Generator().manual_seed(seed)
    losses = []
    for step in range(3):
        x = torch.randn(16, 3, 32, 32, generator=g).bfloat16().cuda()
        x = x.contiguous(memory_format=torch.channels_last)
        y = torch.randint(0, 10, (16,), generator=g).cuda()
        engine.zero_grad()
        loss = F_amd.softmax_cross_entropy(engine(x), y)
        loss.backward()
        engine.finalize_backward()
        opt.step()
        losses.append(loss.float().item())
    os.environ.pop("DLX_IMPL", None)
    return losses


def test_native_vs_torch_impl_loss():
    """Native kernels and the torch fallback must track each other."""
    native_losses = _one_step("auto")
    torch_losses = _one_step("torch")
    for a, b in zip(native_losses, torch_losses):
        assert abs(a - b) < 0.15, (native_losses, torch_losses)
    assert all(torch.isfinite(torch.tensor(native_losses)))


def test_graft_smoke():
    sys.path.insert(0, str(REPO))
    import __graft_entry__ as ge
    ge.smoke()


def test_bench_runs_small():
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--steps", "3",
         "--warmup", "1", "--batch-size", "32"],
        capture_output=True, text=True, timeout=600, cwd=str(REPO), env=env)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 1 and rec["value"] > 0
    assert rec["dtype"] == "bf16"
