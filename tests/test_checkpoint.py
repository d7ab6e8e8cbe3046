"""L7 checkpoint-layout tests (MXNet-compatible naming)."""

import torch

from deeplearning_cfn_amd import checkpoint as ckpt
from deeplearning_cfn_amd.models import build_resnet


def test_layout_and_roundtrip(tmp_path):
    prefix = tmp_path / "ckpt" / "resnet20"
    m = build_resnet(20, dataset="cifar10")
    ckpt.save_symbol(prefix, {"network": "resnet", "num_layers": 20})
    p = ckpt.save_checkpoint(prefix, 3, m)
    assert p.name == "resnet20-0003.params"
    assert (tmp_path / "ckpt" / "resnet20-symbol.json").exists()

    m2 = build_resnet(20, dataset="cifar10")
    state = ckpt.load_checkpoint(prefix, 3, m2)
    assert state["epoch"] == 3
    for a, b in zip(m.parameters(), m2.parameters()):
        torch.testing.assert_close(a, b)
    sym = ckpt.load_symbol(prefix)
    assert sym["num_layers"] == 20


def test_latest_epoch(tmp_path):
    prefix = tmp_path / "resnet50"
    m = torch.nn.Linear(2, 2)
    assert ckpt.latest_epoch(prefix) is None
    ckpt.save_checkpoint(prefix, 1, m)
    ckpt.save_checkpoint(prefix, 12, m)
    assert ckpt.latest_epoch(prefix) == 12


def test_optimizer_state(tmp_path):
    prefix = tmp_path / "m"
    m = torch.nn.Linear(4, 4)
    opt = torch.optim.SGD(m.parameters(), lr=0.1, momentum=0.9)
    m(torch.randn(2, 4)).sum().backward()
    opt.step()
    ckpt.save_checkpoint(prefix, 1, m, opt)
    m2 = torch.nn.Linear(4, 4)
    opt2 = torch.optim.SGD(m2.parameters(), lr=0.5)
    ckpt.load_checkpoint(prefix, 1, m2, opt2)
    assert opt2.state_dict()["param_groups"][0]["lr"] == 0.1
