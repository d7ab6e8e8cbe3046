"""L5 — hand-written CDNA4 HIP ops.

The reference delegates all compute to cuDNN/cuBLAS/framework kernels
(SURVEY.md §3.2); here each delegated hot op is a first-class native
kernel compiled for gfx950 and loaded from the in-tree extension
``_hip_ops``.  Dispatch policy:

* On a GPU (ROCm) box the HIP path is mandatory: a missing extension
  raises unless ``DLX_IMPL=torch`` explicitly selects the fallback.
* On a CPU box (the dev/test container) the pure-torch fallback runs so
  the full engine/model/test stack works without a GPU.
* ``DLX_IMPL`` ∈ {auto (default), native, torch} overrides.
"""

from __future__ import annotations

import os
from pathlib import Path

import torch

_EXT = None
_EXT_ERR: str | None = None


def _try_load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return
    here = Path(__file__).parent
    sos = sorted(here.glob("_hip_ops*.so"))
    if not sos:
        _EXT_ERR = f"no _hip_ops*.so under {here} (build with: python setup.py build_ext --inplace)"
        return
    try:
        torch.ops.load_library(str(sos[0]))
        _EXT = torch.ops.dlx_hip
        _EXT_ERR = None
    except Exception as e:  # pragma: no cover - load failure is env-specific
        _EXT_ERR = f"failed to load {sos[0]}: {e}"


def impl_mode() -> str:
    return os.environ.get("DLX_IMPL", "auto")


def native_available() -> bool:
    if impl_mode() == "torch":
        return False
    if not torch.cuda.is_available():
        return False
    _try_load()
    return _EXT is not None


class _DebugSyncOps:
    """DLX_DEBUG_SYNC=1: synchronize after every native op so an async
    kernel fault (page fault, XID) surfaces at the faulting call site
    instead of a later unrelated one (SURVEY.md §6.2 race/sanitizer
    plan: the cheap device-side invariant)."""

    def __init__(self, ops):
        self._ops = ops

    def __getattr__(self, name):
        fn = getattr(self._ops, name)

        def wrapped(*a, **k):
            r = fn(*a, **k)
            torch.cuda.synchronize()
            return r

        return wrapped


def native():
    """The loaded torch.ops namespace; raises loudly if unavailable on GPU."""
    _try_load()
    if _EXT is None:
        raise RuntimeError(
            f"dlx native HIP extension unavailable: {_EXT_ERR}. "
            f"On a GPU box this is fatal (no silent torch fallback); set "
            f"DLX_IMPL=torch to run the reference-parity fallback explicitly.")
    if os.environ.get("DLX_DEBUG_SYNC") == "1":
        return _DebugSyncOps(_EXT)
    return _EXT


def assert_native_if_gpu() -> None:
    """Fail loudly when a GPU is present but the HIP path can't load."""
    if torch.cuda.is_available() and impl_mode() != "torch":
        native()
