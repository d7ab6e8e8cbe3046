"""``dist_async`` one-step-stale semantics + comm bandwidth metric
(VERDICT r1 #8).

The MXNet ``dist_async`` contract collapsed to DP (engine/ddp.py): step 0
applies the rank's LOCAL gradients; step t>=1 applies the cross-rank
average of step t-1's gradients (one-step stale), while step t's own
gradients are in flight.  The emulated reference below runs both ranks'
replicas in one process and performs that exchange by hand — the same
construction test_train_parity.py uses for ``dist_device_sync``.
"""

import json

import torch

from tests.conftest import run_distributed

from deeplearning_cfn_amd.engine.ddp import DataParallel
from deeplearning_cfn_amd.engine.optim import FlatSGD
from deeplearning_cfn_amd.models import build_resnet

STEPS = 4
LOCAL_B = 4
SEED = 321


def _global_batch(step: int):
    g = torch.Generator().manual_seed(SEED * 1000 + step)
    x = torch.randn(2 * LOCAL_B, 3, 32, 32, generator=g)
    y = torch.randint(0, 10, (2 * LOCAL_B,), generator=g)
    return x, y


def _shard(step: int, rank: int):
    x, y = _global_batch(step)
    return (x[rank * LOCAL_B:(rank + 1) * LOCAL_B],
            y[rank * LOCAL_B:(rank + 1) * LOCAL_B])


def _async_worker(rank, ws, out_path):
    torch.manual_seed(SEED)
    model = build_resnet(20, dataset="cifar10")
    engine = DataParallel(model, mode="dist_async")
    opt = FlatSGD(engine, lr=0.05, momentum=0.9, weight_decay=1e-4)
    losses = []
    for step in range(STEPS):
        x, y = _shard(step, rank)
        engine.zero_grad()
        loss = torch.nn.functional.cross_entropy(engine(x), y)
        loss.backward()
        engine.finalize_backward()
        opt.step()
        losses.append(loss.item())
    # collect both ranks' curves
    t = torch.zeros(ws, STEPS)
    t[rank] = torch.tensor(losses)
    torch.distributed.all_reduce(t)
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump(t.tolist(), f)


def _emulated_async_reference():
    """Both replicas in one process with the staleness exchange by hand."""
    engines, opts = [], []
    for _ in range(2):
        torch.manual_seed(SEED)          # same init == broadcast
        model = build_resnet(20, dataset="cifar10")
        e = DataParallel(model, mode="local")
        engines.append(e)
        opts.append(FlatSGD(e, lr=0.05, momentum=0.9, weight_decay=1e-4))
    losses = [[], []]
    prev = None                           # step t-1's per-rank flat grads
    for step in range(STEPS):
        cur = []
        for r in range(2):
            x, y = _shard(step, r)
            engines[r].zero_grad()
            loss = torch.nn.functional.cross_entropy(engines[r](x), y)
            loss.backward()
            losses[r].append(loss.item())
            cur.append([b.flat.clone() for b in engines[r].plan.buckets])
        if prev is not None:
            # consume the one-step-stale cross-rank average
            for i in range(len(cur[0])):
                avg = (prev[0][i] + prev[1][i]) / 2
                for r in range(2):
                    engines[r].plan.buckets[i].flat.copy_(avg)
        # step 0: local grads stay in place (PS pull-before-first-push)
        for r in range(2):
            opts[r].step()
        prev = cur
    return losses


def test_dist_async_one_step_stale_parity(tmp_path):
    out = tmp_path / "async_losses.json"
    run_distributed(_async_worker, 2, str(out))
    dist_losses = json.load(open(out))
    ref = _emulated_async_reference()
    for r in range(2):
        for a, b in zip(dist_losses[r], ref[r]):
            assert abs(a - b) < 1e-4, (dist_losses, ref)
    # staleness must actually matter: curves differ from a fully-sync run
    assert any(abs(dist_losses[0][i] - dist_losses[1][i]) > 1e-7
               for i in range(1, STEPS))


def _bw_worker(rank, ws, out_path):
    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(256, 256),
                                torch.nn.ReLU(),
                                torch.nn.Linear(256, 64))
    engine = DataParallel(model, bucket_bytes=64 << 10)
    rows = engine.comm_bandwidth_probe(iters=3)
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump(rows, f)


def test_comm_bandwidth_probe_ws2(tmp_path):
    out = tmp_path / "bw.json"
    run_distributed(_bw_worker, 2, str(out))
    rows = json.load(open(out))
    assert len(rows) >= 2                 # bucketed into several buckets
    for r in rows:
        assert r["mb"] > 0 and r["ms"] > 0 and r["busbw_gbps"] > 0
