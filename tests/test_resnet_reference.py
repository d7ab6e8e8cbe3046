"""Model-wiring parity: our ResNet vs an independent plain-torch twin.

The fused block forwards (conv_bn_act) hardcode strides/paddings per
position; this test would catch any mis-wiring by comparing logits and
gradients against a straightforward torch-only implementation with the
same weights (CPU, fp32 — both sides use torch ops there, so equality
is near-exact).
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from deeplearning_cfn_amd.models import build_resnet


def _twin_bottleneck(x, blk, training):
    identity = x
    if blk.downsample is not None:
        d = blk.downsample
        identity = F.batch_norm(
            F.conv2d(x, d.conv.weight, None, d.conv.stride[0], 0),
            d.bn.running_mean, d.bn.running_var, d.bn.weight, d.bn.bias,
            training, d.bn.momentum, d.bn.eps)
    out = F.relu(F.batch_norm(
        F.conv2d(x, blk.conv1.weight, None, 1, 0),
        blk.bn1.bn.running_mean, blk.bn1.bn.running_var, blk.bn1.bn.weight,
        blk.bn1.bn.bias, training, 0.1, 1e-5))
    out = F.relu(F.batch_norm(
        F.conv2d(out, blk.conv2.weight, None, blk.conv2.stride[0], 1),
        blk.bn2.bn.running_mean, blk.bn2.bn.running_var, blk.bn2.bn.weight,
        blk.bn2.bn.bias, training, 0.1, 1e-5))
    out = F.batch_norm(
        F.conv2d(out, blk.conv3.weight, None, 1, 0),
        blk.bn3.bn.running_mean, blk.bn3.bn.running_var, blk.bn3.bn.weight,
        blk.bn3.bn.bias, training, 0.1, 1e-5)
    return F.relu(out + identity)


def test_resnet50_twin_forward_backward():
    torch.manual_seed(0)
    model = build_resnet(50)
    model.train()

    x = torch.randn(2, 3, 64, 64, requires_grad=True)
    y = model(x)

    # twin forward with the same parameters, independent wiring
    m2 = build_resnet(50)
    m2.load_state_dict(model.state_dict())
    m2.train()
    x2 = x.detach().clone().requires_grad_()
    h = F.relu(F.batch_norm(
        F.conv2d(x2, m2.conv1.weight, None, 2, 3),
        m2.bn1.bn.running_mean, m2.bn1.bn.running_var, m2.bn1.bn.weight,
        m2.bn1.bn.bias, True, 0.1, 1e-5))
    h = F.max_pool2d(h, 3, stride=2, padding=1)
    for stage in m2.stages:
        for blk in stage:
            h = _twin_bottleneck(h, blk, True)
    h = F.adaptive_avg_pool2d(h, 1)
    y2 = m2.fc(torch.flatten(h, 1))

    torch.testing.assert_close(y, y2, rtol=1e-4, atol=1e-4)

    g = torch.randn_like(y)
    y.backward(g)
    y2.backward(g)
    torch.testing.assert_close(x.grad, x2.grad, rtol=1e-3, atol=1e-4)
    p1 = dict(model.named_parameters())
    p2 = dict(m2.named_parameters())
    for name in ["stages.0.0.conv2.weight", "stages.1.0.downsample.conv.weight",
                 "stages.3.2.conv3.weight", "stages.0.0.bn1.bn.weight"]:
        torch.testing.assert_close(p1[name].grad, p2[name].grad,
                                   rtol=1e-3, atol=1e-3)
