"""L6 model-zoo tests: shapes, parameter counts, CLI mapping."""

import pytest
import torch

from deeplearning_cfn_amd.models import build_model, build_resnet, build_bert


def test_resnet20_cifar_shapes():
    m = build_resnet(20, dataset="cifar10")
    y = m(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)
    n = sum(p.numel() for p in m.parameters())
    assert 0.25e6 < n < 0.30e6  # ResNet-20 ~0.27M params


def test_resnet50_imagenet_shapes():
    m = build_resnet(50)
    y = m(torch.randn(1, 3, 224, 224))
    assert y.shape == (1, 1000)
    n = sum(p.numel() for p in m.parameters())
    assert 25.0e6 < n < 26.0e6  # ResNet-50 ~25.56M


def test_resnet152_builds():
    m = build_resnet(152)
    n = sum(p.numel() for p in m.parameters())
    assert 59e6 < n < 61e6  # ResNet-152 ~60.2M


def test_bad_depths():
    with pytest.raises(ValueError):
        build_resnet(21, dataset="cifar10")
    with pytest.raises(ValueError):
        build_resnet(77)


def test_bert_base_forward():
    m = build_bert("base", seq_len=32)
    ids = torch.randint(0, 30522, (2, 32))
    labels = ids.clone()
    labels[:, ::2] = -100
    nsp = torch.randint(0, 2, (2,))
    loss = m(ids, labels, nsp)
    assert loss.dim() == 0 and torch.isfinite(loss)
    n = sum(p.numel() for p in m.parameters())
    assert 105e6 < n < 115e6  # BERT-base ~110M


def test_registry():
    m = build_model("resnet", num_layers=18)
    assert sum(p.numel() for p in m.parameters()) > 1e7
    with pytest.raises(ValueError):
        build_model("vgg")


def test_resnet_backward_channels_last():
    m = build_resnet(18).to(memory_format=torch.channels_last)
    x = torch.randn(2, 3, 224, 224).contiguous(
        memory_format=torch.channels_last)
    m(x).sum().backward()
    for p in m.parameters():
        assert p.grad is not None
