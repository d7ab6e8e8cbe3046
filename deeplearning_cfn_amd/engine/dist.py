"""L1 — membership & rendezvous.

The reference discovers membership via an SQS handshake and a hostfile
(SURVEY.md §2 L1/§4.1).  Here membership is the launcher's env (RANK /
WORLD_SIZE / MASTER_*), and the handshake is torch.distributed's TCP store
rendezvous.  Backend selection:

* ``nccl`` — RCCL over xGMI on ROCm; one process per GPU.
* ``gloo`` — CPU path used by the no-GPU parity config
  (BASELINE.json config #1) and all multi-rank unit tests.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist


@dataclass
class DistContext:
    rank: int
    local_rank: int
    world_size: int
    backend: str
    device: torch.device

    @property
    def is_master(self) -> bool:
        return self.rank == 0

    def barrier(self) -> None:
        if dist.is_initialized() and self.world_size > 1:
            dist.barrier()


def init_from_env(backend: Optional[str] = None,
                  timeout_s: int = 600) -> DistContext:
    """Initialise the process group from the launcher's env contract.

    Safe to call in a single-process run with no env set (world of 1, no
    process group) — mirrors running the reference's training script
    without ``launch.py``/``mpirun``.
    """
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    world_size = int(os.environ.get("WORLD_SIZE", "1"))

    cuda = torch.cuda.is_available()
    if backend is None:
        backend = "nccl" if cuda else "gloo"
    if backend == "nccl" and not cuda:
        raise RuntimeError(
            "backend nccl (RCCL) requested but no GPU is visible; use gloo "
            "for CPU runs")

    if cuda:
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
        device = torch.device("cuda", torch.cuda.current_device())
    else:
        device = torch.device("cpu")

    if world_size > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world_size,
            timeout=datetime.timedelta(seconds=timeout_s))
    return DistContext(rank=rank, local_rank=local_rank,
                       world_size=world_size, backend=backend, device=device)


def shutdown() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()
