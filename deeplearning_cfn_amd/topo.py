"""L0 residue — node topology preflight.

The reference's CloudFormation template provisions the cluster and its
WaitCondition asserts every worker checked in.
On one MI355X node that collapses to: enumerate the GPUs, sanity-check the
xGMI mesh assumptions the engine tunes for, and fail fast with a clear
message — the moral equivalent of a stack rollback.

xGMI facts the engine's bucket sizing uses:
8 GPUs full mesh, 7 point-to-point links per GPU, ~153 GB/s per link.
"""

from __future__ import annotations

import json
import os
import shutil
import subprocess
from dataclasses import dataclass, field
from typing import List

# Tuning constants for the MI355X node.
XGMI_LINKS_PER_GPU = 7
XGMI_GBPS_PER_LINK = 153.0
HBM_BYTES_PER_GPU = 288 * (1 << 30)
CUS_PER_GPU = 256
XCDS_PER_GPU = 8


@dataclass
class NodeTopology:
    n_gpus: int
    gpu_names: List[str] = field(default_factory=list)
    hbm_bytes: int = HBM_BYTES_PER_GPU
    xgmi_links_per_gpu: int = XGMI_LINKS_PER_GPU
    xgmi_gbps_per_link: float = XGMI_GBPS_PER_LINK

    @property
    def is_full_mesh(self) -> bool:
        return self.n_gpus <= self.xgmi_links_per_gpu + 1

    def to_json(self) -> str:
        return json.dumps(self.__dict__, indent=2)


def probe(require_gpu: bool = False) -> NodeTopology:
    """Enumerate visible GPUs.

    Uses torch (HIP runtime) when available; falls back to rocm-smi; on a
    GPU-less dev box returns an 0-GPU topology unless ``require_gpu``.
    """
    import torch

    if torch.cuda.is_available():
        n = torch.cuda.device_count()
        names = [torch.cuda.get_device_name(i) for i in range(n)]
        props = torch.cuda.get_device_properties(0)
        return NodeTopology(n_gpus=n, gpu_names=names,
                            hbm_bytes=int(props.total_memory))
    # rocm-smi fallback (covers the case torch was built CPU-only)
    smi = shutil.which("rocm-smi")
    if smi:
        try:
            out = subprocess.run([smi, "--showid", "--json"],
                                 capture_output=True, text=True, timeout=30)
            if out.returncode == 0 and out.stdout.strip():
                cards = [k for k in json.loads(out.stdout) if k.startswith("card")]
                if cards:
                    return NodeTopology(n_gpus=len(cards),
                                        gpu_names=["AMD Instinct"] * len(cards))
        except Exception:
            pass
    if require_gpu:
        raise RuntimeError(
            "preflight: no AMD GPU visible (torch.cuda.is_available() is "
            "False and rocm-smi found none). On the dev container this is "
            "expected; run via gpurun on an MI355X box.")
    return NodeTopology(n_gpus=0)


def preflight(num_ranks: int, require_gpu: bool = True) -> NodeTopology:
    """Validate the box can host ``num_ranks`` ranks (1 rank per GPU).

    Mirrors the reference master's "drain SQS until WorkerCount check-ins"
    loop: here the check-in is simply device visibility.
    """
    topo = probe(require_gpu=require_gpu)
    if require_gpu and topo.n_gpus < num_ranks:
        raise RuntimeError(
            f"preflight: {num_ranks} ranks requested but only {topo.n_gpus} "
            f"GPUs visible (HIP_VISIBLE_DEVICES={os.environ.get('HIP_VISIBLE_DEVICES')!r})")
    return topo
