"""L7 — checkpoint/resume in the reference-compatible layout.

The MXNet example writes ``{prefix}-symbol.json`` (architecture manifest)
plus ``{prefix}-{epoch:04d}.params`` per epoch on shared EFS and resumes
via ``--load-epoch N``.  We keep that layout in spirit:

* ``{prefix}-symbol.json`` — a JSON manifest: model family, constructor
  args, dtype, world size, framework version (written once by rank 0);
* ``{prefix}-{epoch:04d}.params`` — a torch-serialised dict of
  ``{"model": state_dict, "optimizer": state_dict, "epoch": int, ...}``;
* resume: all ranks barrier, rank 0's file is loaded everywhere (the
  files are on the shared ``$EFS_MOUNT`` so every rank sees them), then
  parameters broadcast from rank 0 for belt-and-braces determinism.
"""

from __future__ import annotations

import json
import re
from pathlib import Path
from typing import Any, Dict, Optional

import torch


def symbol_path(prefix: str | Path) -> Path:
    return Path(f"{prefix}-symbol.json")


def params_path(prefix: str | Path, epoch: int) -> Path:
    return Path(f"{prefix}-{epoch:04d}.params")


def save_symbol(prefix: str | Path, manifest: Dict[str, Any]) -> None:
    p = symbol_path(prefix)
    p.parent.mkdir(parents=True, exist_ok=True)
    p.write_text(json.dumps(manifest, indent=2, sort_keys=True))


def load_symbol(prefix: str | Path) -> Dict[str, Any]:
    return json.loads(symbol_path(prefix).read_text())


def save_checkpoint(
    prefix: str | Path,
    epoch: int,
    model: torch.nn.Module,
    optimizer: Optional[torch.optim.Optimizer] = None,
    extra: Optional[Dict[str, Any]] = None,
) -> Path:
    """Rank-0-only in multi-rank runs (caller gates on rank)."""
    p = params_path(prefix, epoch)
    p.parent.mkdir(parents=True, exist_ok=True)
    state = {
        "model": model.state_dict(),
        "optimizer": optimizer.state_dict() if optimizer else None,
        "epoch": epoch,
    }
    if extra:
        state.update(extra)
    tmp = p.with_suffix(".params.tmp")
    torch.save(state, tmp)
    tmp.replace(p)  # atomic publish so a crashed writer never truncates
    return p


def load_checkpoint(
    prefix: str | Path,
    epoch: int,
    model: torch.nn.Module,
    optimizer: Optional[torch.optim.Optimizer] = None,
    map_location: str = "cpu",
) -> Dict[str, Any]:
    state = torch.load(params_path(prefix, epoch), map_location=map_location,
                       weights_only=False)
    model.load_state_dict(state["model"])
    if optimizer is not None:
        # Flat optimizers capture fp32 masters at construction; loading the
        # model copies new values into the flat param buffers, so masters
        # must be re-derived first.  Checkpointed masters (if saved) then
        # overwrite these with the exact fp32 values.
        if hasattr(optimizer, "refresh_masters"):
            optimizer.refresh_masters()
        if state.get("optimizer") is not None:
            optimizer.load_state_dict(state["optimizer"])
    return state


def latest_epoch(prefix: str | Path) -> Optional[int]:
    """Newest ``{prefix}-NNNN.params`` on disk (elastic auto-resume)."""
    prefix = Path(prefix)
    pat = re.compile(re.escape(prefix.name) + r"-(\d{4})\.params$")
    best = None
    if prefix.parent.exists():
        for f in prefix.parent.iterdir():
            m = pat.match(f.name)
            if m:
                e = int(m.group(1))
                best = e if best is None else max(best, e)
    return best
