"""Worker for tests/test_rccl_gpu.py — one rank of a 2-rank RCCL world.

Both ranks are pinned to the same physical GPU (HIP_VISIBLE_DEVICES=0),
which exercises the real RCCL communicator/bucketed-collective path on
the 1-GPU lease.  Modes (env DLX_RCCL_MODE):

sync   — engine dist_device_sync parity: each rank trains its shard of a
         deterministic global batch; rank 0 also computes the emulated
         single-process reference and writes both loss curves.
async  — dist_async double-buffer cycling smoke.
graph  — hipGraph capture of {compute + all_reduce}; replay must match
         an eager execution of the same sequence.
"""

import json
import os
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

SEED = 777
STEPS = 4
LOCAL_B = 4


def _init():
    rank = int(os.environ["RANK"])
    try:
        dist.init_process_group("nccl", rank=rank, world_size=2)
        # first collective is where a duplicate-device refusal surfaces
        probe = torch.ones(8, device="cuda")
        dist.all_reduce(probe)
        torch.cuda.synchronize()
        assert probe[0].item() == 2.0
    except Exception as e:  # noqa: BLE001
        print(f"DUP_GPU_REFUSED rank={rank}: {type(e).__name__}: {e}",
              flush=True)
        sys.exit(3)
    return rank


def _global_batch(step):
    g = torch.Generator().manual_seed(SEED * 1000 + step)
    x = torch.randn(2 * LOCAL_B, 3, 32, 32, generator=g)
    y = torch.randint(0, 10, (2 * LOCAL_B,), generator=g)
    return x, y


def _build(mode):
    from deeplearning_cfn_amd.engine.ddp import DataParallel
    from deeplearning_cfn_amd.engine.optim import FlatSGD
    from deeplearning_cfn_amd.models import build_resnet
    torch.manual_seed(SEED)
    model = build_resnet(20, dataset="cifar10").cuda()
    engine = DataParallel(model, mode=mode)
    opt = FlatSGD(engine, lr=0.05, momentum=0.9, weight_decay=1e-4)
    return engine, opt


def run_sync(rank, out_dir):
    engine, opt = _build("dist_device_sync")
    losses = []
    for step in range(STEPS):
        x, y = _global_batch(step)
        xs = x[rank * LOCAL_B:(rank + 1) * LOCAL_B].cuda()
        ys = y[rank * LOCAL_B:(rank + 1) * LOCAL_B].cuda()
        engine.zero_grad()
        loss = torch.nn.functional.cross_entropy(engine(xs), ys)
        loss.backward()
        engine.finalize_backward()
        opt.step()
        losses.append(loss.item())
    t = torch.tensor(losses, device="cuda")
    dist.all_reduce(t)
    t /= 2
    if rank == 0:
        # emulated single-process reference (test_train_parity.py shape)
        from deeplearning_cfn_amd.engine.ddp import DataParallel
        from deeplearning_cfn_amd.engine.optim import FlatSGD
        from deeplearning_cfn_amd.models import build_resnet
        torch.manual_seed(SEED)
        model = build_resnet(20, dataset="cifar10").cuda()
        e2 = DataParallel(model, mode="local")
        o2 = FlatSGD(e2, lr=0.05, momentum=0.9, weight_decay=1e-4)
        ref = []
        for step in range(STEPS):
            x, y = _global_batch(step)
            e2.zero_grad()
            sl = []
            for r in range(2):
                xs = x[r * LOCAL_B:(r + 1) * LOCAL_B].cuda()
                ys = y[r * LOCAL_B:(r + 1) * LOCAL_B].cuda()
                loss = torch.nn.functional.cross_entropy(e2(xs), ys)
                loss.backward()
                sl.append(loss.item())
            for b in e2.plan.buckets:
                b.flat.mul_(0.5)
            o2.step()
            ref.append(sum(sl) / 2)
        (Path(out_dir) / "rank0.json").write_text(json.dumps(
            {"dp_losses": t.tolist(), "ref_losses": ref}))


def run_async(rank, out_dir):
    engine, opt = _build("dist_async")
    losses = []
    for step in range(STEPS + 2):
        x, y = _global_batch(step)
        xs = x[rank * LOCAL_B:(rank + 1) * LOCAL_B].cuda()
        ys = y[rank * LOCAL_B:(rank + 1) * LOCAL_B].cuda()
        engine.zero_grad()
        loss = torch.nn.functional.cross_entropy(engine(xs), ys)
        loss.backward()
        engine.finalize_backward()
        opt.step()
        losses.append(loss.item())
    if rank == 0:
        (Path(out_dir) / "rank0.json").write_text(json.dumps(
            {"losses": losses}))


def run_graph(rank, out_dir):
    """Capture {axpy kernel + all_reduce + scale} in a hipGraph, replay
    3x, compare against eager execution of the identical sequence."""
    n = 1 << 20
    x = torch.full((n,), float(rank + 1), device="cuda")
    buf = torch.empty_like(x)

    def step_fn():
        buf.copy_(x * 2.0)
        dist.all_reduce(buf)
        buf.mul_(0.5)

    # eager reference
    eager = []
    for _ in range(3):
        step_fn()
        eager.append(buf.clone())
    torch.cuda.synchronize()

    err = None
    try:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                step_fn()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            step_fn()
    except Exception as e:  # noqa: BLE001
        err = e
    # agree on capture success BEFORE replaying (a lone replayer would
    # hang in its graphed collective)
    flag = torch.tensor([0.0 if err is None else 1.0], device="cuda")
    dist.all_reduce(flag)
    torch.cuda.synchronize()
    if flag.item() != 0.0:
        print(f"GRAPH_CAPTURE_UNSUPPORTED rank={rank}: "
              f"{type(err).__name__ if err else 'peer'}: {err}", flush=True)
        dist.barrier()
        sys.exit(0)

    ok = True
    for _ in range(3):
        g.replay()
        torch.cuda.synchronize()
        # each rank contributes rank+1 doubled: sum = 2*(1+2), avg = 3
        ok = ok and torch.allclose(buf, torch.full_like(buf, 3.0))
    if rank == 0:
        (Path(out_dir) / "rank0.json").write_text(json.dumps(
            {"graph_matches_eager": bool(ok)}))
    dist.barrier()


def main():
    mode = os.environ["DLX_RCCL_MODE"]
    out_dir = os.environ["DLX_RCCL_OUT"]
    rank = _init()
    torch.cuda.set_device(0)
    try:
        {"sync": run_sync, "async": run_async, "graph": run_graph}[mode](
            rank, out_dir)
        dist.barrier()
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
