"""End-to-end convergence (GPU): the whole native stack must LEARN.

Memorization of a fixed 4-batch synthetic pool drives CIFAR ResNet-20
from ln(10)≈2.30 to ~0.01 in 400 steps (measured curve in
profiles/convergence_runs.log) — a numerics bug in any kernel of the
conv/BN/SGD chain breaks this.  (Endless fresh random labels have
ln(C) as their optimum, hence the pool.)
"""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_cifar_memorization_converges():
    from deeplearning_cfn_amd import train as train_mod

    args = train_mod.build_argparser().parse_args([
        "--network", "resnet", "--num-layers", "20", "--dataset", "cifar10",
        "--batch-size", "256", "--steps-per-epoch", "400",
        "--disp-batches", "100", "--num-epochs", "1",
        "--checkpoint-every", "0", "--lr", "0.05", "--data-pool", "4",
    ])
    out = train_mod.train(args)
    assert out["final_loss"] < 0.5, (
        f"native stack failed to memorize: final loss {out['final_loss']}")
