"""Metrics / logging / observability.

The reference surfaces: interleaved rank stdout, MXNet per-epoch
``Validation-accuracy`` lines, and CloudWatch defaults.  Rebuild:
rank-0 console lines with images/sec and per-bucket all-reduce bandwidth
(vs the 153 GB/s/link xGMI ceiling), plus a JSONL metrics file under
``$EFS_MOUNT`` for harnesses to parse.
"""

from __future__ import annotations

import json
import time
from pathlib import Path
from typing import Any, Dict, Optional


class MetricsLogger:
    def __init__(self, path: Optional[str | Path] = None, rank: int = 0,
                 quiet: bool = False):
        self.rank = rank
        self.quiet = quiet
        self.path = Path(path) if path else None
        self._fh = None
        if self.path and rank == 0:
            self.path.parent.mkdir(parents=True, exist_ok=True)
            self._fh = open(self.path, "a", buffering=1)

    def log(self, kind: str, **fields: Any) -> None:
        rec: Dict[str, Any] = {"ts": time.time(), "kind": kind, **fields}
        if self._fh:
            self._fh.write(json.dumps(rec) + "\n")
        if self.rank == 0 and not self.quiet:
            msg = " ".join(f"{k}={v:.4g}" if isinstance(v, float) else f"{k}={v}"
                           for k, v in fields.items())
            print(f"[{kind}] {msg}", flush=True)

    def close(self) -> None:
        if self._fh:
            self._fh.close()


class StepTimer:
    """Throughput accounting; device-synchronised when on GPU."""

    def __init__(self, device):
        self.device = device
        self.t0 = None

    def start(self) -> None:
        self._sync()
        self.t0 = time.perf_counter()

    def stop(self) -> float:
        self._sync()
        dt = time.perf_counter() - self.t0
        self.t0 = None
        return dt

    def _sync(self) -> None:
        import torch
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
