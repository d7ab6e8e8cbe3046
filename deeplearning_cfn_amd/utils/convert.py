"""Mixed-precision policy: compute dtype everywhere, BatchNorm fp32.

bf16 activations/weights with fp32 normalization statistics is the
numerically sound MXNet/Horovod-era recipe and
the contract of the native BN kernels (x bf16/fp32, params fp32).
"""

from __future__ import annotations

import torch
from torch.nn.modules.batchnorm import _BatchNorm


def to_compute_dtype(model: torch.nn.Module,
                     dtype: torch.dtype) -> torch.nn.Module:
    if dtype == torch.float32:
        return model.float()
    model.to(dtype)
    for m in model.modules():
        if isinstance(m, (_BatchNorm, torch.nn.LayerNorm)):
            m.float()
    return model
