"""BERT through the DP engine on gloo world_size=2 (config #4 semantics).

Mirrors test_train_parity for the transformer path: the 2-rank loss
curve (with FlatAdam) must match an emulated single-process data-parallel
reference — covering the engine's bucketing/collectives over the BERT
parameter set (embeddings, LayerNorm fp32 params interleaved with
linear weights) before the driver's multi-GPU runs.
"""

import json

import torch

from tests.conftest import run_distributed

from deeplearning_cfn_amd.engine.ddp import DataParallel
from deeplearning_cfn_amd.engine.optim import FlatAdam
from deeplearning_cfn_amd.models.bert import BertConfig, BertForPretraining

STEPS = 3
LOCAL_B = 2
SEED = 77


def _tiny_bert():
    cfg = BertConfig(vocab_size=211, hidden=64, layers=2, heads=4,
                     intermediate=128, max_seq=32)
    return BertForPretraining(cfg)


def _global_batch(step: int):
    g = torch.Generator().manual_seed(SEED * 991 + step)
    ids = torch.randint(0, 211, (2 * LOCAL_B, 16), generator=g)
    labels = ids.clone()
    mask = torch.rand(ids.shape, generator=g) < 0.3
    labels[~mask] = -100
    nsp = torch.randint(0, 2, (2 * LOCAL_B,), generator=g)
    return ids, labels, nsp


def _step_losses(engine, opt, rank, ws):
    losses = []
    for step in range(STEPS):
        ids, labels, nsp = _global_batch(step)
        if ws > 1:
            sl = slice(rank * LOCAL_B, (rank + 1) * LOCAL_B)
            ids, labels, nsp = ids[sl], labels[sl], nsp[sl]
        engine.zero_grad()
        loss = engine(ids, labels, nsp)
        loss.backward()
        engine.finalize_backward()
        opt.step()
        losses.append(loss.item())
    return losses


def _dp_worker(rank, ws, out_path):
    torch.manual_seed(SEED)
    model = _tiny_bert()
    engine = DataParallel(model, mode="dist_device_sync")
    opt = FlatAdam(engine, lr=1e-3, weight_decay=0.01)
    losses = _step_losses(engine, opt, rank, ws)
    t = torch.tensor(losses)
    torch.distributed.all_reduce(t)
    t /= ws
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump(t.tolist(), f)


def _emulated_reference():
    torch.manual_seed(SEED)
    model = _tiny_bert()
    engine = DataParallel(model, mode="local")
    opt = FlatAdam(engine, lr=1e-3, weight_decay=0.01)
    losses = []
    for step in range(STEPS):
        ids, labels, nsp = _global_batch(step)
        engine.zero_grad()
        shard = []
        for r in range(2):
            sl = slice(r * LOCAL_B, (r + 1) * LOCAL_B)
            loss = engine(ids[sl], labels[sl], nsp[sl])
            loss.backward()
            shard.append(loss.item())
        for b in engine.plan.buckets:
            b.flat.mul_(0.5)
        opt.step()
        losses.append(sum(shard) / 2)
    return losses


def test_bert_dp_loss_parity_ws2(tmp_path):
    out = tmp_path / "bert_losses.json"
    run_distributed(_dp_worker, 2, str(out))
    dp = json.load(open(out))
    ref = _emulated_reference()
    for a, b in zip(dp, ref):
        assert abs(a - b) < 2e-4, (dp, ref)
    assert dp[-1] != dp[0]
