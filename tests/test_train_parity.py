"""BASELINE config #1 — the MXNet-example parity milestone.

CIFAR-10 ResNet-20 data-parallel on CPU/gloo world_size=2: the loss curve
of the distributed run must match a single-process run on the identical
global batch (fixed seed), proving launcher + engine + bucketing +
collective semantics end-to-end without a GPU.
"""

import json

import torch

from tests.conftest import run_distributed

from deeplearning_cfn_amd.engine.ddp import DataParallel
from deeplearning_cfn_amd.engine.optim import FlatSGD
from deeplearning_cfn_amd.models import build_resnet

STEPS = 4
LOCAL_B = 4
SEED = 123


def _global_batch(step: int):
    g = torch.Generator().manual_seed(SEED * 1000 + step)
    x = torch.randn(2 * LOCAL_B, 3, 32, 32, generator=g)
    y = torch.randint(0, 10, (2 * LOCAL_B,), generator=g)
    return x, y


def _train(model, engine, opt, rank, ws):
    losses = []
    for step in range(STEPS):
        x, y = _global_batch(step)
        if ws > 1:
            x = x[rank * LOCAL_B:(rank + 1) * LOCAL_B]
            y = y[rank * LOCAL_B:(rank + 1) * LOCAL_B]
        engine.zero_grad()
        logits = engine(x)
        loss = torch.nn.functional.cross_entropy(logits, y)
        loss.backward()
        engine.finalize_backward()
        opt.step()
        losses.append(loss.item())
    return losses


def _dp_worker(rank, ws, out_path):
    torch.manual_seed(SEED)
    model = build_resnet(20, dataset="cifar10")
    engine = DataParallel(model, mode="dist_device_sync")
    opt = FlatSGD(engine, lr=0.05, momentum=0.9, weight_decay=1e-4)
    losses = _train(model, engine, opt, rank, ws)
    # global loss per step = mean over ranks (equal local batch sizes)
    t = torch.tensor(losses)
    torch.distributed.all_reduce(t)
    t /= ws
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump(t.tolist(), f)


def _emulated_dp_reference():
    """Single process doing both ranks' work: per-shard forward/backward
    (BatchNorm sees the LOCAL shard, exactly as each rank does), gradient
    sum / world_size, one optimizer step — the mathematical definition of
    the collective the engine implements."""
    torch.manual_seed(SEED)
    model = build_resnet(20, dataset="cifar10")
    engine = DataParallel(model, mode="local")
    opt = FlatSGD(engine, lr=0.05, momentum=0.9, weight_decay=1e-4)
    losses = []
    for step in range(STEPS):
        x, y = _global_batch(step)
        engine.zero_grad()
        shard_losses = []
        for r in range(2):
            xs = x[r * LOCAL_B:(r + 1) * LOCAL_B]
            ys = y[r * LOCAL_B:(r + 1) * LOCAL_B]
            loss = torch.nn.functional.cross_entropy(engine(xs), ys)
            loss.backward()  # accumulates into the flat buckets
            shard_losses.append(loss.item())
        for b in engine.plan.buckets:
            b.flat.mul_(0.5)
        opt.step()
        losses.append(sum(shard_losses) / 2)
    return losses


def test_dp_loss_parity_ws2(tmp_path):
    out = tmp_path / "dp_losses.json"
    run_distributed(_dp_worker, 2, str(out))
    dp_losses = json.load(open(out))
    ref_losses = _emulated_dp_reference()
    for a, b in zip(dp_losses, ref_losses):
        assert abs(a - b) < 1e-4, (dp_losses, ref_losses)
    # training must actually move
    assert ref_losses[-1] != ref_losses[0]
