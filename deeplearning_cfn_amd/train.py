"""Per-rank training entry point — ``train_cifar10.py`` CLI parity.

The MXNet example's flags are accepted verbatim and mapped to the RCCL
engine: ``--network resnet --num-layers N
--batch-size B --kv-store dist_device_sync --gpus 0,...,7 --model-prefix P
--load-epoch N --lr --mom --wd --num-epochs``.  Additional flags cover
the rebuild's knobs: ``--dtype bf16``, ``--impl native|torch|auto``,
``--bucket-mb``, ``--synthetic`` shapes, ``--profile``.

Run standalone (1 process) or under the launcher / torch.distributed.run
(one process per GPU; env contract + rendezvous vars respected).
"""

from __future__ import annotations

import argparse
import os
import sys

from pathlib import Path

import torch

from . import checkpoint as ckpt
from .contract import EFS_MOUNT
from .data import HostPinnedPrefetcher, SyntheticBertBatches, SyntheticImages
from .engine import DataParallel, init_from_env
from .engine.dist import shutdown
from .engine.kvstore import resolve as resolve_kvstore
from .engine.optim import FlatAdam, FlatSGD
from .metrics import MetricsLogger, StepTimer
from .models import build_model
from .models.registry import model_manifest
from .ops import assert_native_if_gpu


def build_argparser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        description="deeplearning_cfn_amd trainer (MXNet-example CLI parity)")
    # --- reference-compatible flags ---
    p.add_argument("--network", default="resnet", help="resnet | bert")
    p.add_argument("--num-layers", type=int, default=50)
    p.add_argument("--batch-size", type=int, default=128,
                   help="per-rank batch size")
    p.add_argument("--kv-store", default="dist_device_sync",
                   help="dist_device_sync | dist_sync | dist_async | device | local")
    p.add_argument("--gpus", default=None,
                   help="comma list, e.g. 0,1,2,3 (sets HIP visibility order)")
    p.add_argument("--model-prefix", default=None,
                   help="checkpoint prefix (default $EFS_MOUNT/checkpoints/<network>)")
    p.add_argument("--load-epoch", type=int, default=None)
    p.add_argument("--num-epochs", type=int, default=1)
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--mom", type=float, default=0.9)
    p.add_argument("--wd", type=float, default=1e-4)
    p.add_argument("--disp-batches", type=int, default=20)
    # --- rebuild knobs ---
    p.add_argument("--dataset", default="imagenet",
                   help="imagenet | cifar10 (synthetic shapes)")
    p.add_argument("--dtype", default="bf16", help="bf16 | fp32")
    p.add_argument("--impl", default="auto", help="auto | native | torch")
    p.add_argument("--bucket-mb", type=float, default=None,
                   help="gradient bucket size; default auto (xGMI-sized)")
    p.add_argument("--steps-per-epoch", type=int, default=100)
    p.add_argument("--seq-len", type=int, default=128, help="bert only")
    p.add_argument("--seed", type=int, default=42)
    p.add_argument("--no-prefetch", action="store_true")
    p.add_argument("--data-pool", type=int, default=0,
                   help="cycle a fixed pool of N synthetic batches "
                        "(0 = endless fresh batches); a finite pool makes "
                        "the loss decrease (memorization) for convergence "
                        "checks")
    p.add_argument("--checkpoint-every", type=int, default=1,
                   help="epochs between checkpoints (0 = off)")
    p.add_argument("--no-auto-resume", action="store_true",
                   help="do not resume from the newest checkpoint under "
                        "--model-prefix when --load-epoch is absent "
                        "(default: resume — the reference's elastic-"
                        "relaunch semantics, SURVEY.md §6.3/§4.4)")
    p.add_argument("--eval-steps", type=int, default=4,
                   help="validation batches per epoch (0 = no eval); "
                        "logged as the MXNet-style Validation-accuracy "
                        "line (SURVEY.md §6.5)")
    p.add_argument("--fault-step", type=int, default=None,
                   help="test-only: exit(13) at this step on DLX_FAULT_RANK")
    p.add_argument("--profile", action="store_true",
                   help="capture a per-op trace of --profile-steps steps to "
                        "$EFS_MOUNT/trace_rank<r>.json (chrome://tracing); "
                        "for kernel-level counters run under rocprofv3")
    p.add_argument("--profile-steps", type=int, default=10)
    return p


def _dtype(name: str) -> torch.dtype:
    return {"bf16": torch.bfloat16, "bfloat16": torch.bfloat16,
            "fp32": torch.float32, "float32": torch.float32}[name]


@torch.no_grad()
def _evaluate(engine, eval_batches, args, ctx) -> float:
    """Per-epoch validation pass — the MXNet example's per-epoch
    ``Validation-accuracy`` analogue.

    ResNet path: top-1 accuracy; BERT path: masked-token prediction
    accuracy.  Correct/total counts are all-reduced so the printed number
    is the global validation accuracy, as in the reference's logs.
    """
    model = engine.module
    was_training = model.training
    model.eval()
    correct = total = 0
    for _ in range(args.eval_steps):
        batch = next(eval_batches)
        if args.network == "bert":
            ids, mlm, _ = (t.to(ctx.device) for t in batch)
            logits = model(ids)                     # [B,S,V], no labels
            mask = mlm != -100
            pred = logits.argmax(dim=-1)
            correct += (pred[mask] == mlm[mask]).sum().item()
            total += int(mask.sum().item())
        else:
            x, y = batch
            x, y = x.to(ctx.device), y.to(ctx.device)
            if ctx.device.type == "cuda":
                x = x.contiguous(memory_format=torch.channels_last)
            pred = model(x).argmax(dim=-1)
            correct += (pred == y).sum().item()
            total += y.numel()
    if was_training:
        model.train()
    import torch.distributed as dist
    if ctx.world_size > 1 and dist.is_initialized():
        t = torch.tensor([correct, total], dtype=torch.float64)
        dist.all_reduce(t)
        correct, total = t.tolist()
    return correct / max(total, 1)


def train(args: argparse.Namespace) -> dict:
    if args.impl != "auto":
        os.environ["DLX_IMPL"] = {"native": "auto", "torch": "torch",
                                  "auto": "auto"}[args.impl]
    if args.gpus and "HIP_VISIBLE_DEVICES" not in os.environ:
        os.environ.setdefault("CUDA_VISIBLE_DEVICES", args.gpus)

    ctx = init_from_env()
    assert_native_if_gpu()
    dtype = _dtype(args.dtype) if ctx.device.type == "cuda" else torch.float32
    mode = resolve_kvstore(args.kv_store)
    torch.manual_seed(args.seed)  # same init on every rank pre-broadcast

    is_bert = args.network == "bert"
    model_kw = (dict(size="base", seq_len=args.seq_len) if is_bert else
                dict(num_layers=args.num_layers, dataset=args.dataset))
    model = build_model(args.network, **model_kw)
    model = model.to(ctx.device)
    if not is_bert:
        model = model.to(memory_format=torch.channels_last)
    from .utils import to_compute_dtype
    model = to_compute_dtype(model, dtype)

    bucket_bytes = int(args.bucket_mb * (1 << 20)) if args.bucket_mb else None
    engine = DataParallel(model, bucket_bytes=bucket_bytes, mode=mode)
    if is_bert:
        opt = FlatAdam(engine, lr=args.lr, weight_decay=args.wd)
    else:
        opt = FlatSGD(engine, lr=args.lr, momentum=args.mom,
                      weight_decay=args.wd)

    efs = os.environ.get(EFS_MOUNT, "/tmp/dlx_efs")
    prefix = args.model_prefix or str(
        Path(efs) / "checkpoints" / f"{args.network}{args.num_layers}")
    logger = MetricsLogger(Path(efs) / "metrics.jsonl", rank=ctx.rank)

    start_epoch = 0
    load_epoch = args.load_epoch
    if (load_epoch is None and not args.no_auto_resume
            and args.checkpoint_every):
        # Elastic relaunch resumes from the last shared-mount checkpoint
        # (reference semantics): every rank sees the
        # same $EFS_MOUNT, so latest_epoch() agrees across ranks.
        load_epoch = ckpt.latest_epoch(prefix)
    if load_epoch is not None:
        state = ckpt.load_checkpoint(prefix, load_epoch, model, opt,
                                     map_location=str(ctx.device))
        start_epoch = state["epoch"]
        engine.broadcast_parameters()
        logger.log("resume", epoch=start_epoch, prefix=prefix)
    elif ctx.is_master:
        ckpt.save_symbol(prefix, model_manifest(args.network, **model_kw))

    # data
    if is_bert:
        source = SyntheticBertBatches(args.batch_size, seq_len=args.seq_len,
                                      seed=args.seed, rank=ctx.rank,
                                      pinned=ctx.device.type == "cuda",
                                      pool=args.data_pool)
    else:
        image_size = 32 if args.dataset.startswith("cifar") else 224
        ncls = 10 if args.dataset.startswith("cifar") else 1000
        source = SyntheticImages(args.batch_size, image_size=image_size,
                                 num_classes=ncls, seed=args.seed,
                                 rank=ctx.rank, dtype=dtype,
                                 channels_last=True,
                                 pinned=ctx.device.type == "cuda",
                                 pool=args.data_pool)
    prefetcher = None
    if ctx.device.type == "cuda" and not args.no_prefetch:
        prefetcher = HostPinnedPrefetcher(source, ctx.device)
        batches = iter(prefetcher)
    else:
        batches = iter(source)

    # validation source: with a finite memorization pool, "validation" is
    # the pool itself (held-out random labels are chance forever — the
    # synthetic analogue of the CIFAR example's accuracy is memorization
    # accuracy); endless-data runs get a held-out seed.
    eval_batches = None
    if args.eval_steps:
        if args.data_pool > 0:
            eval_batches = iter(source)
        elif is_bert:
            eval_batches = iter(SyntheticBertBatches(
                args.batch_size, seq_len=args.seq_len,
                seed=args.seed + 7919, rank=ctx.rank))
        else:
            eval_batches = iter(SyntheticImages(
                args.batch_size, image_size=image_size, num_classes=ncls,
                seed=args.seed + 7919, rank=ctx.rank, dtype=dtype,
                channels_last=True))

    if ctx.world_size > 1 and args.kv_store != "local":
        # per-bucket all-reduce bandwidth vs the 153 GB/s xGMI link
        # ceiling (SURVEY.md §6.5 metrics contract)
        from .topo import XGMI_GBPS_PER_LINK
        for row in engine.comm_bandwidth_probe():
            logger.log("comm_bw", bucket=row["bucket"], mb=row["mb"],
                       ms=row["ms"], busbw_gbps=row["busbw_gbps"],
                       link_ceiling_gbps=XGMI_GBPS_PER_LINK)

    from .ops import functional as F_amd
    fault_rank = int(os.environ.get("DLX_FAULT_RANK", "-1"))
    timer = StepTimer(ctx.device)
    global_step = 0
    losses: list[float] = []

    profiler = None
    if args.profile:
        # tracing is first-class.  torch.profiler rides
        # roctracer on ROCm; the chrome trace lands on the shared mount so
        # every rank's capture is inspectable from the master.
        acts = [torch.profiler.ProfilerActivity.CPU]
        if ctx.device.type == "cuda":
            acts.append(torch.profiler.ProfilerActivity.CUDA)
        trace_path = str(Path(efs) / f"trace_rank{ctx.rank}.json")
        profiler = torch.profiler.profile(
            activities=acts,
            schedule=torch.profiler.schedule(
                wait=1, warmup=2, active=args.profile_steps, repeat=1),
            on_trace_ready=lambda p: p.export_chrome_trace(trace_path))
        profiler.__enter__()
        logger.log("profile", path=trace_path, steps=args.profile_steps)

    for epoch in range(start_epoch, args.num_epochs):
        model.train()
        timer.start()
        interval_samples = 0
        for step in range(args.steps_per_epoch):
            batch = next(batches)
            engine.zero_grad()
            if is_bert:
                ids, mlm, nsp = (t.to(ctx.device) for t in batch)
                loss = engine(ids, mlm, nsp)
            else:
                x, y = batch
                x, y = x.to(ctx.device), y.to(ctx.device)
                if ctx.device.type == "cuda":
                    x = x.contiguous(memory_format=torch.channels_last)
                logits = engine(x)
                loss = F_amd.softmax_cross_entropy(logits, y)
            loss.backward()
            engine.finalize_backward()
            opt.step()
            if profiler is not None:
                profiler.step()
            global_step += 1
            interval_samples += args.batch_size
            if (args.fault_step is not None and ctx.rank == fault_rank
                    and global_step >= args.fault_step):
                print(f"[fault-injection] rank {ctx.rank} exiting at step "
                      f"{global_step}", flush=True)
                sys.stdout.flush()
                os._exit(13)
            if (step + 1) % args.disp_batches == 0:
                dt = timer.stop()
                ips = interval_samples * ctx.world_size / dt
                losses.append(loss.item())
                logger.log("train", epoch=epoch, step=step + 1,
                           loss=losses[-1], samples_per_sec=ips)
                timer.start()
                interval_samples = 0
        timer.stop()
        if args.eval_steps:
            acc = _evaluate(engine, eval_batches, args, ctx)
            if ctx.rank == 0:
                # the MXNet example's per-epoch line, verbatim shape
                # (SURVEY.md §6.5): "Epoch[N] Validation-accuracy=V"
                print(f"Epoch[{epoch}] Validation-accuracy={acc:.6f}",
                      flush=True)
            logger.log("eval", epoch=epoch, validation_accuracy=acc)
        if (ctx.is_master and args.checkpoint_every
                and (epoch + 1) % args.checkpoint_every == 0):
            ckpt.save_checkpoint(prefix, epoch + 1, model, opt)
            logger.log("checkpoint", epoch=epoch + 1, prefix=prefix)
        ctx.barrier()

    if profiler is not None:
        profiler.__exit__(None, None, None)
    if prefetcher is not None:
        prefetcher.close()
    final_loss = losses[-1] if losses else float("nan")
    logger.log("done", final_loss=final_loss, steps=global_step)
    logger.close()
    return {"final_loss": final_loss, "losses": losses, "steps": global_step}


def main(argv=None) -> int:
    args = build_argparser().parse_args(argv)
    try:
        train(args)
    finally:
        shutdown()
    return 0


if __name__ == "__main__":
    sys.exit(main())
