"""MXNet-kvstore parity shim (SURVEY.md §3.2 PS-Lite row, §6.6).

The MXNet example selects its distribution strategy with
``--kv-store dist_device_sync`` (also ``dist_sync`` / ``dist_async`` /
``device`` / ``local``).  Those names are accepted here and mapped onto
the RCCL DP engine:

========================  =============================================
reference kvstore          engine behaviour
========================  =============================================
``dist_device_sync``       synchronous bucketed all-reduce, aggregation
                           on device (the default; == Horovod parity)
``dist_sync``              same collective semantics (host-aggregated in
                           MXNet; aggregation locus is moot on one node)
``dist_async``             one-step-stale asynchronous exchange
``device`` / ``local``     no collective (single-process)
========================  =============================================
"""

from __future__ import annotations

from typing import Optional

import torch

from .ddp import DataParallel

_ALIASES = {
    "dist_device_sync": "dist_device_sync",
    "dist_sync": "dist_sync",
    "dist_async": "dist_async",
    "device": "local",
    "local": "local",
}


def resolve(name: str) -> str:
    try:
        return _ALIASES[name]
    except KeyError:
        raise ValueError(
            f"unknown --kv-store {name!r}; accepted: {sorted(_ALIASES)}"
        ) from None


def create(name: str = "dist_device_sync"):
    """``mxnet.kvstore.create``-shaped factory returning a wrapper fn.

    Usage parity:  ``kv = create('dist_device_sync'); model = kv(model)``.
    """
    mode = resolve(name)

    def wrap(module: torch.nn.Module,
             bucket_bytes: Optional[int] = None) -> DataParallel:
        return DataParallel(module, bucket_bytes=bucket_bytes, mode=mode)

    wrap.mode = mode
    wrap.type = name
    return wrap
