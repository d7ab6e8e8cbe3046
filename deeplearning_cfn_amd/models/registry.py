"""Model registry: maps the reference's CLI surface to constructors.

``--network resnet --num-layers N`` (MXNet example flags)
and ``--network bert`` (BASELINE config #4) resolve here.
"""

from __future__ import annotations

from typing import Any, Dict

import torch

from .resnet import build_resnet
from .bert import build_bert

MODEL_FAMILIES = ("resnet", "bert")


def build_model(network: str, **kw: Any) -> torch.nn.Module:
    if network == "resnet":
        return build_resnet(
            num_layers=int(kw.get("num_layers", 50)),
            num_classes=kw.get("num_classes"),
            dataset=kw.get("dataset", "imagenet"),
        )
    if network == "bert":
        return build_bert(size=kw.get("size", "base"),
                          seq_len=int(kw.get("seq_len", 128)))
    raise ValueError(f"unknown --network {network!r}; families: {MODEL_FAMILIES}")


def model_manifest(network: str, **kw: Any) -> Dict[str, Any]:
    """The ``-symbol.json`` content (checkpoint.py layout)."""
    return {"network": network, "args": kw,
            "framework": "deeplearning_cfn_amd"}
