"""L2 — the environment-variable contract.

The reference installs these variables in ``/etc/profile.d`` on the master
during stack creation (SURVEY.md §2 L2, §3.1 "Master boot script"); every
launcher and example recipe consumes them.  The names are kept verbatim:

``DEEPLEARNING_WORKERS_PATH``
    Path to the hostfile, one rank-host per line.  On a single MI355X node
    the launcher writes one line per rank (``localhost slots=1``-style
    entries reduced to plain hostnames).
``DEEPLEARNING_WORKERS_COUNT``
    Number of entries in the hostfile (= number of training ranks).
``DEEPLEARNING_WORKER_GPU_COUNT``
    GPUs visible per host (8 on a full MI355X node).
``EFS_MOUNT``
    Shared data/checkpoint directory.  The reference mounts EFS here; we
    default to a local directory with identical semantics (checkpoints,
    datasets, code visible to every rank).
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from pathlib import Path
from typing import Mapping, Optional

WORKERS_PATH = "DEEPLEARNING_WORKERS_PATH"
WORKERS_COUNT = "DEEPLEARNING_WORKERS_COUNT"
WORKER_GPU_COUNT = "DEEPLEARNING_WORKER_GPU_COUNT"
EFS_MOUNT = "EFS_MOUNT"

#: Variables the launcher additionally exports per rank (torch.distributed
#: rendezvous names; RCCL == the "nccl" backend on ROCm).
RANK = "RANK"
LOCAL_RANK = "LOCAL_RANK"
WORLD_SIZE = "WORLD_SIZE"
MASTER_ADDR = "MASTER_ADDR"
MASTER_PORT = "MASTER_PORT"


@dataclass(frozen=True)
class EnvContract:
    """The resolved contract, as a value object."""

    workers_path: str
    workers_count: int
    worker_gpu_count: int
    efs_mount: str

    def export(self, env: Optional[dict] = None) -> dict:
        """Write the contract into ``env`` (default: a copy of os.environ)."""
        out = dict(os.environ) if env is None else env
        out[WORKERS_PATH] = self.workers_path
        out[WORKERS_COUNT] = str(self.workers_count)
        out[WORKER_GPU_COUNT] = str(self.worker_gpu_count)
        out[EFS_MOUNT] = self.efs_mount
        return out

    @staticmethod
    def from_env(env: Optional[Mapping[str, str]] = None) -> "EnvContract":
        e = os.environ if env is None else env
        try:
            return EnvContract(
                workers_path=e[WORKERS_PATH],
                workers_count=int(e[WORKERS_COUNT]),
                worker_gpu_count=int(e[WORKER_GPU_COUNT]),
                efs_mount=e[EFS_MOUNT],
            )
        except KeyError as k:
            raise KeyError(
                f"env contract incomplete: {k} not set; run `dlx up` or the "
                f"launcher first (see docs/recipes.md)"
            ) from None


def write_hostfile(path: str | Path, hosts: list[str]) -> None:
    """Write the workers file, one host per line.

    Mirrors the reference master's SQS-drain loop that appends one private
    IP per check-in to ``$DEEPLEARNING_WORKERS_PATH``.
    """
    p = Path(path)
    p.parent.mkdir(parents=True, exist_ok=True)
    p.write_text("".join(h + "\n" for h in hosts))


def read_hostfile(path: str | Path) -> list[str]:
    """Read the workers file, ignoring blanks/comments (launcher input)."""
    out = []
    for line in Path(path).read_text().splitlines():
        line = line.split("#", 1)[0].strip()
        if line:
            # accept "host" or "host slots=N" (mpirun-hostfile style)
            out.append(line.split()[0])
    return out


def make_contract(
    num_ranks: int,
    gpus_per_host: int,
    efs_mount: str | Path,
    hostfile: Optional[str | Path] = None,
) -> EnvContract:
    """Build (and write) the single-node contract.

    The reference resolves COUNT = number of hostfile entries; we define one
    entry per rank on localhost (SURVEY.md §9 appendix ruling).
    """
    efs = Path(efs_mount)
    efs.mkdir(parents=True, exist_ok=True)
    hf = Path(hostfile) if hostfile is not None else efs / "workers"
    write_hostfile(hf, ["localhost"] * num_ranks)
    return EnvContract(
        workers_path=str(hf),
        workers_count=num_ranks,
        worker_gpu_count=gpus_per_host,
        efs_mount=str(efs),
    )
