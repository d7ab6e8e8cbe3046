"""Dispatching functional layer: HIP-native on GPU, torch fallback on CPU.

Each function is the seam the north star names: the torch path is the
correctness reference (tests compare native vs a plain fp32 torch
implementation of the same op), the HIP path is the production one.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from . import native, native_available


def _use_native(x: torch.Tensor) -> bool:
    return x.is_cuda and native_available()


def _to_nhwc(x: torch.Tensor) -> torch.Tensor:
    return x.contiguous(memory_format=torch.channels_last)


# ---------------------------------------------------------------------------
# fused residual add + ReLU (tail of every ResNet block)
# ---------------------------------------------------------------------------

class _AddReLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
        out = native().add_relu_fwd(a, b)
        ctx.save_for_backward(out)
        return out

    @staticmethod
    def backward(ctx, grad: torch.Tensor):
        (out,) = ctx.saved_tensors
        g = native().relu_bwd(grad, out)  # op contiguates to out's layout
        return g, g


def add_relu(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    if _use_native(a) and a.dtype in (torch.bfloat16, torch.float16, torch.float32):
        return _AddReLU.apply(a, b)
    return F.relu(a + b)


# ---------------------------------------------------------------------------
# standalone ReLU (kept for completeness with the reference's op list)
# ---------------------------------------------------------------------------

class _ReLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor) -> torch.Tensor:
        out = native().relu_fwd(x)
        ctx.save_for_backward(out)
        return out

    @staticmethod
    def backward(ctx, grad: torch.Tensor):
        (out,) = ctx.saved_tensors
        return native().relu_bwd(grad, out)


def relu(x: torch.Tensor) -> torch.Tensor:
    if _use_native(x):
        return _ReLU.apply(x)
    return F.relu(x)


# ---------------------------------------------------------------------------
# BatchNorm (+ fused ReLU) — NHWC-native, fp32 statistics
# ---------------------------------------------------------------------------

class _BNAct(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var,
                momentum: float, eps: float, relu_: bool):
        x = _to_nhwc(x)
        y, save_mean, save_invstd, mask = native().bn_fwd_train(
            x, weight, bias, running_mean, running_var, momentum, eps, relu_)
        # with a relu bit-mask, backward never re-reads y (2 fewer passes)
        if mask.numel():
            ctx.save_for_backward(x, weight, save_mean, save_invstd, mask)
            ctx.masked = True
        else:
            ctx.save_for_backward(x, weight, save_mean, save_invstd, y)
            ctx.masked = False
        ctx.relu = relu_
        return y

    @staticmethod
    def backward(ctx, grad: torch.Tensor):
        x, weight, save_mean, save_invstd, aux = ctx.saved_tensors
        if ctx.masked:
            dx, dweight, dbias = native().bn_bwd_m(
                _to_nhwc(grad), x, weight, save_mean, save_invstd, aux)
        else:
            dx, dweight, dbias = native().bn_bwd(
                _to_nhwc(grad), x, weight, save_mean, save_invstd, aux,
                ctx.relu)
        return dx, dweight, dbias, None, None, None, None, None


def batch_norm_act(x: torch.Tensor, bn: torch.nn.BatchNorm2d,
                   relu: bool = True, training: bool = True) -> torch.Tensor:
    """BatchNorm2d forward with optional fused ReLU.

    Native path: one NHWC HIP kernel pair (Welford-style fp32 stats +
    normalize+activate fused — replaces cuDNN BN + framework ReLU,
    SURVEY.md §3.2 rows cuDNN-BatchNorm / framework-elementwise).
    """
    if training and _use_native(x):
        return _BNAct.apply(x, bn.weight, bn.bias, bn.running_mean,
                            bn.running_var, bn.momentum, bn.eps, relu)
    if not training and _use_native(x):
        return native().bn_fwd_eval(_to_nhwc(x), bn.weight, bn.bias,
                                    bn.running_mean, bn.running_var,
                                    bn.eps, relu)
    out = F.batch_norm(x, bn.running_mean, bn.running_var, bn.weight,
                       bn.bias, training, bn.momentum, bn.eps)
    return F.relu(out) if relu else out


# ---------------------------------------------------------------------------
# fused BatchNorm + residual add + ReLU (ResNet block tail)
# ---------------------------------------------------------------------------

class _BNAddReLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, residual, weight, bias, running_mean, running_var,
                momentum: float, eps: float):
        x = _to_nhwc(x)
        y, save_mean, save_invstd, mask = native().bn_fwd_train_res(
            x, weight, bias, running_mean, running_var, momentum, eps,
            _to_nhwc(residual))
        ctx.save_for_backward(x, weight, save_mean, save_invstd, mask)
        return y

    @staticmethod
    def backward(ctx, grad: torch.Tensor):
        x, weight, save_mean, save_invstd, mask = ctx.saved_tensors
        dx, dweight, dbias, dres = native().bn_bwd_res_m(
            _to_nhwc(grad), x, weight, save_mean, save_invstd, mask)
        return dx, dres, dweight, dbias, None, None, None, None


def batch_norm_add_relu(x: torch.Tensor, bn: torch.nn.BatchNorm2d,
                        residual: torch.Tensor,
                        training: bool = True) -> torch.Tensor:
    """relu(batch_norm(x) + residual) — the ResNet block tail, fused.

    Native path: BN stats reduce + one normalize/add/relu pass; backward
    emits dx AND the relu-masked residual grad in a single pass
    (replaces cuDNN BN + framework add + ReLU).
    """
    if _use_native(x) and x.shape[1] % (8 if x.dtype == torch.bfloat16
                                        else 4) == 0:
        if training:
            return _BNAddReLU.apply(x, residual, bn.weight, bn.bias,
                                    bn.running_mean, bn.running_var,
                                    bn.momentum, bn.eps)
        return native().bn_fwd_eval_res(_to_nhwc(x), bn.weight, bn.bias,
                                        bn.running_mean, bn.running_var,
                                        bn.eps, _to_nhwc(residual))
    out = F.batch_norm(x, bn.running_mean, bn.running_var, bn.weight,
                       bn.bias, training, bn.momentum, bn.eps)
    return F.relu(out + residual)


# ---------------------------------------------------------------------------
# fully fused conv + BatchNorm (+ReLU / +residual) — the ResNet layer unit
# ---------------------------------------------------------------------------

class _ConvBNAct(torch.autograd.Function):
    """conv -> BN (+relu / +residual+relu) with the BN statistics reduced
    in the conv epilogue (conv2d_fwd_stats): the whole-tensor BN reduce
    pass disappears.  Backward chains bn_bwd(_res)_m then conv dgrad/wrw.
    """

    @staticmethod
    def forward(ctx, x, weight, stride: int, pad: int, bnw, bnb,
                running_mean, running_var, momentum: float, eps: float,
                relu_: bool, residual):
        x = _to_nhwc(x)
        yc, sums = native().conv2d_fwd_stats(x, weight, stride, pad)
        if residual is not None:
            y, mean, invstd, mask = native().bn_fwd_train_res_presum(
                yc, bnw, bnb, running_mean, running_var, momentum, eps,
                _to_nhwc(residual), sums)
        else:
            y, mean, invstd, mask = native().bn_fwd_train_presum(
                yc, bnw, bnb, running_mean, running_var, momentum, eps,
                relu_, sums)
        ctx.save_for_backward(x, weight, yc, bnw, mean, invstd, mask)
        ctx.meta = (stride, pad, relu_, residual is not None)
        return y

    @staticmethod
    def backward(ctx, grad: torch.Tensor):
        x, weight, yc, bnw, mean, invstd, mask = ctx.saved_tensors
        s, p, relu_, has_res = ctx.meta
        grad = _to_nhwc(grad)
        dres = None
        if has_res:
            dyc, dgamma, dbeta, dres = native().bn_bwd_res_m(
                grad, yc, bnw, mean, invstd, mask)
        elif relu_:
            dyc, dgamma, dbeta = native().bn_bwd_m(
                grad, yc, bnw, mean, invstd, mask)
        else:
            # relu=False never reads its y argument
            dyc, dgamma, dbeta = native().bn_bwd(
                grad, yc, bnw, mean, invstd, yc, False)
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dx = native().conv2d_dgrad(dyc, weight, x.shape[2], x.shape[3],
                                       s, p)
        if ctx.needs_input_grad[1]:
            if _os.environ.get("DLX_CONV_WRW", "native") == "native":
                dw = native().conv2d_wrw(dyc, x, weight.shape[2],
                                         weight.shape[3], s, p)
            else:
                dw = torch.ops.aten.convolution_backward(
                    dyc, x, weight, None, [s, s], [p, p], [1, 1], False,
                    [0, 0], 1, [False, True, False])[1]
        return (dx, dw, None, None, dgamma, dbeta, None, None, None, None,
                None, dres)


def conv_bn_act(x: torch.Tensor, weight: torch.Tensor,
                bn: torch.nn.BatchNorm2d, stride: int = 1, padding: int = 0,
                relu: bool = True, residual: Optional[torch.Tensor] = None,
                training: bool = True) -> torch.Tensor:
    """The ResNet layer unit: bn(conv(x)) (+residual) (+relu), fused.

    Native path computes the BN channel sums inside the conv epilogue
    (SURVEY.md §3.2 conv/BN rows collapsed into one producer).  Falls
    back to the composable ops outside the native gate.
    """
    if (training and _use_native(x) and _conv_policy() == "native"
            and x.dtype == torch.bfloat16
            and (weight.shape[1] % 8 == 0 or weight.shape[1] <= 4)
            and weight.shape[0] % 8 == 0):
        return _ConvBNAct.apply(x, weight, stride, padding, bn.weight,
                                bn.bias, bn.running_mean, bn.running_var,
                                bn.momentum, bn.eps, relu, residual)
    out = conv2d(x, weight, stride, padding)
    if residual is not None:
        return batch_norm_add_relu(out, bn, residual, training)
    return batch_norm_act(out, bn, relu=relu, training=training)


# ---------------------------------------------------------------------------
# whole-bottleneck fusion (ResNet-50/101/152 block) — VERDICT r1 weak #2:
# the autograd fan-in at the residual junction costs one big at::add per
# block; hand-chaining the block backward lands that add in conv1's
# dgrad epilogue (conv2d_dgrad_acc).
# ---------------------------------------------------------------------------


class _BottleneckFn(torch.autograd.Function):
    """conv1/bn1+relu -> conv2/bn2+relu -> conv3/bn3(+res)+relu with an
    optional conv/bn downsample branch, trained end-to-end through the
    native kernels.  Forward matches the composed _ConvBNAct chain
    op-for-op; backward chains bn_bwd_*_m / dgrad / wrw by hand and
    fuses the residual fan-in into dgrad_acc."""

    @staticmethod
    def forward(ctx, x, w1, w2, w3, wd,
                g1, b1, g2, b2, g3, b3, gd, bd,
                rm1, rv1, rm2, rv2, rm3, rv3, rmd, rvd,
                s2, p2, sd, momentum, eps):
        n = native()
        x = _to_nhwc(x)
        yc1, sums1 = n.conv2d_fwd_stats(x, w1, 1, 0)
        y1, m1, i1, k1 = n.bn_fwd_train_presum(yc1, g1, b1, rm1, rv1,
                                               momentum, eps, True, sums1)
        yc2, sums2 = n.conv2d_fwd_stats(y1, w2, s2, p2)
        y2, m2, i2, k2 = n.bn_fwd_train_presum(yc2, g2, b2, rm2, rv2,
                                               momentum, eps, True, sums2)
        yc3, sums3 = n.conv2d_fwd_stats(y2, w3, 1, 0)
        if wd is not None:
            ycd, sumsd = n.conv2d_fwd_stats(x, wd, sd, 0)
            res, md, idd, _ = n.bn_fwd_train_presum(
                ycd, gd, bd, rmd, rvd, momentum, eps, False, sumsd)
        else:
            ycd = md = idd = None
            res = x
        y3, m3, i3, k3 = n.bn_fwd_train_res_presum(
            yc3, g3, b3, rm3, rv3, momentum, eps, res, sums3)
        ctx.save_for_backward(x, w1, w2, w3, wd, g1, g2, g3, gd,
                              yc1, yc2, yc3, ycd, y1, y2,
                              m1, i1, k1, m2, i2, k2, m3, i3, k3, md, idd)
        ctx.meta = (s2, p2, sd)
        return y3

    @staticmethod
    def backward(ctx, grad):
        n = native()
        (x, w1, w2, w3, wd, g1, g2, g3, gd,
         yc1, yc2, yc3, ycd, y1, y2,
         m1, i1, k1, m2, i2, k2, m3, i3, k3, md, idd) = ctx.saved_tensors
        s2, p2, sd = ctx.meta
        g = _to_nhwc(grad)
        dyc3, dg3, db3, dres = n.bn_bwd_res_m(g, yc3, g3, m3, i3, k3)
        dy2 = n.conv2d_dgrad(dyc3, w3, y2.shape[2], y2.shape[3], 1, 0)
        dw3 = n.conv2d_wrw(dyc3, y2, 1, 1, 1, 0)
        dyc2, dg2, db2 = n.bn_bwd_m(dy2, yc2, g2, m2, i2, k2)
        dy1 = n.conv2d_dgrad(dyc2, w2, y1.shape[2], y1.shape[3], s2, p2)
        dw2 = n.conv2d_wrw(dyc2, y1, w2.shape[2], w2.shape[3], s2, p2)
        dyc1, dg1, db1 = n.bn_bwd_m(dy1, yc1, g1, m1, i1, k1)
        dw1 = n.conv2d_wrw(dyc1, x, 1, 1, 1, 0)
        dwd = dgd = dbd = None
        if wd is not None:
            # bn (no relu) backward never reads its y argument
            dycd, dgd, dbd = n.bn_bwd(dres, ycd, gd, md, idd, ycd, False)
            accum = n.conv2d_dgrad(dycd, wd, x.shape[2], x.shape[3], sd, 0)
            dwd = n.conv2d_wrw(dycd, x, 1, 1, sd, 0)
        else:
            accum = dres
        # the fused residual fan-in: dx = conv1-dgrad + accum in one pass
        dx = n.conv2d_dgrad_acc(dyc1, w1, x.shape[2], x.shape[3], 1, 0,
                                accum)
        return (dx, dw1, dw2, dw3, dwd, dg1, db1, dg2, db2, dg3, db3,
                dgd, dbd) + (None,) * 13


def bottleneck(x, conv1w, bn1, conv2w, bn2, stride2, pad2, conv3w, bn3,
               down_convw=None, down_bn=None, down_stride=1,
               training=True):
    """The full ResNet bottleneck unit through one autograd node."""
    # Measured (profiles/r2_experiments/r2f.log, b2048): the one-node block is ~1.4%
    # SLOWER than the composed chain despite eliminating the fan-in
    # at::add (and graphs replay it worse still) — default OFF, kept as
    # an opt-in with its parity test (tests/test_block_fusion_gpu.py).
    if (training and _use_native(x) and _conv_policy() == "native"
            and x.dtype == torch.bfloat16 and x.shape[1] % 8 == 0
            and _os.environ.get("DLX_BLOCK_FUSION", "0") == "1"):
        gd = bd = rmd = rvd = None
        if down_bn is not None:
            gd, bd = down_bn.weight, down_bn.bias
            rmd, rvd = down_bn.running_mean, down_bn.running_var
        return _BottleneckFn.apply(
            x, conv1w, conv2w, conv3w, down_convw,
            bn1.weight, bn1.bias, bn2.weight, bn2.bias,
            bn3.weight, bn3.bias, gd, bd,
            bn1.running_mean, bn1.running_var,
            bn2.running_mean, bn2.running_var,
            bn3.running_mean, bn3.running_var, rmd, rvd,
            stride2, pad2, down_stride, bn1.momentum, bn1.eps)
    # composed fallback (CPU / eval / non-native)
    identity = x if down_convw is None else conv_bn_act(
        x, down_convw, down_bn, down_stride, 0, relu=False,
        training=training)
    out = conv_bn_act(x, conv1w, bn1, 1, 0, relu=True, training=training)
    out = conv_bn_act(out, conv2w, bn2, stride2, pad2, relu=True,
                      training=training)
    return conv_bn_act(out, conv3w, bn3, 1, 0, residual=identity,
                       training=training)


# ---------------------------------------------------------------------------
# fused softmax + cross-entropy (the classifier head)
# ---------------------------------------------------------------------------

class _SoftmaxCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, target: torch.Tensor):
        loss, probs = native().softmax_ce_fwd(logits.contiguous(), target)
        ctx.save_for_backward(probs, target)
        ctx.out_bf16 = logits.dtype == torch.bfloat16
        return loss

    @staticmethod
    def backward(ctx, grad: torch.Tensor):
        probs, target = ctx.saved_tensors
        return native().softmax_ce_bwd(probs, target, grad, ctx.out_bf16), None


def softmax_cross_entropy(logits: torch.Tensor,
                          target: torch.Tensor) -> torch.Tensor:
    """Mean cross-entropy over the batch (fused softmax+CE on HIP)."""
    if _use_native(logits) and logits.dim() == 2:
        return _SoftmaxCE.apply(logits, target)
    return F.cross_entropy(logits.float(), target)


# ---------------------------------------------------------------------------
# layout pack (NHWC <-> NCHW) — explicit north-star op
# ---------------------------------------------------------------------------

def nchw_to_nhwc(x: torch.Tensor) -> torch.Tensor:
    if _use_native(x) and x.dim() == 4:
        return native().pack_nchw_to_nhwc(x)
    return x.contiguous(memory_format=torch.channels_last)


def nhwc_to_nchw(x: torch.Tensor) -> torch.Tensor:
    if _use_native(x) and x.dim() == 4:
        return native().pack_nhwc_to_nchw(x)
    return x.contiguous(memory_format=torch.contiguous_format)


# ---------------------------------------------------------------------------
# convolution — implicit-GEMM MFMA (fwd + dgrad native; wgrad library)
# ---------------------------------------------------------------------------

import os as _os


def _conv_policy() -> str:
    return _os.environ.get("DLX_CONV", "native")


class _Conv2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, stride: int, pad: int):
        y = native().conv2d_fwd(x, weight, stride, pad)
        ctx.save_for_backward(x, weight)
        ctx.sp = (stride, pad)
        return y

    @staticmethod
    def backward(ctx, grad):
        x, w = ctx.saved_tensors
        s, p = ctx.sp
        grad = grad.contiguous(memory_format=torch.channels_last)
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dx = native().conv2d_dgrad(grad, w, x.shape[2], x.shape[3], s, p)
        if ctx.needs_input_grad[1]:
            if _os.environ.get("DLX_CONV_WRW", "native") == "native":
                # native MFMA wrw (m-reduction GEMM via ds_read_b64_tr_b16)
                dw = native().conv2d_wrw(grad, x, w.shape[2], w.shape[3], s, p)
            else:
                dw = torch.ops.aten.convolution_backward(
                    grad, x, w, None, [s, s], [p, p], [1, 1], False, [0, 0],
                    1, [False, True, False])[1]
        return dx, dw, None, None


def conv2d(x: torch.Tensor, weight: torch.Tensor, stride: int = 1,
           padding: int = 0) -> torch.Tensor:
    """2D conv, no bias/dilation/groups (the ResNet surface).

    Native implicit-GEMM when: GPU + bf16 + Cin%8==0 + Cout%8==0 and the
    policy allows; otherwise the library path (stem conv Cin=3 included).
    """
    if (_use_native(x) and _conv_policy() == "native"
            and x.dtype == torch.bfloat16
            and (weight.shape[1] % 8 == 0 or weight.shape[1] <= 4)
            and weight.shape[0] % 8 == 0):
        return _Conv2d.apply(x, weight, stride, padding)
    return F.conv2d(x, weight, None, stride, padding)


# ---------------------------------------------------------------------------
# pooling (ResNet stem maxpool 3x3/s2/p1, global average pool)
# ---------------------------------------------------------------------------

class _MaxPool3x3s2(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        y, argmax = native().maxpool3x3s2_fwd(x)
        ctx.save_for_backward(argmax)
        ctx.hw = (x.shape[2], x.shape[3])
        return y

    @staticmethod
    def backward(ctx, grad):
        (argmax,) = ctx.saved_tensors
        return native().maxpool3x3s2_bwd(grad, argmax, *ctx.hw)


def maxpool3x3s2(x: torch.Tensor) -> torch.Tensor:
    if _use_native(x):
        return _MaxPool3x3s2.apply(x)
    return F.max_pool2d(x, 3, stride=2, padding=1)


class _GlobalAvgPool(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.hw = (x.shape[2], x.shape[3])
        return native().gap_fwd(x)

    @staticmethod
    def backward(ctx, grad):
        return native().gap_bwd(grad, *ctx.hw)


def global_avg_pool(x: torch.Tensor) -> torch.Tensor:
    if _use_native(x):
        return _GlobalAvgPool.apply(x)
    return F.adaptive_avg_pool2d(x, 1)


# ---------------------------------------------------------------------------
# LayerNorm + GELU (BERT path)
# ---------------------------------------------------------------------------

class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        y, mean, invstd = native().ln_fwd(x, weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, invstd)
        return y

    @staticmethod
    def backward(ctx, grad):
        x, weight, mean, invstd = ctx.saved_tensors
        dx, dw, db = native().ln_bwd(grad, x, weight, mean, invstd)
        return dx, dw, db, None


def layer_norm(x, normalized_shape, weight, bias, eps=1e-5):
    if (_use_native(x) and weight is not None
            and weight.dtype == torch.float32 and len(normalized_shape) == 1):
        return _LayerNorm.apply(x, weight, bias, eps)
    return F.layer_norm(x, normalized_shape, weight, bias, eps)


class _GELU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        return native().gelu_fwd(x)

    @staticmethod
    def backward(ctx, grad):
        (x,) = ctx.saved_tensors
        return native().gelu_bwd(grad, x)


def gelu(x: torch.Tensor) -> torch.Tensor:
    if _use_native(x):
        return _GELU.apply(x)
    return F.gelu(x)


# ---------------------------------------------------------------------------
# GEMM / Linear (MFMA kernel; used by the FC head and the BERT path)
# ---------------------------------------------------------------------------

class _LinearMFMA(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        y = native().gemm_nt(x, weight)  # [M,K] @ [N,K]^T -> [M,N]
        if bias is not None:
            y = y + bias
        return y

    @staticmethod
    def backward(ctx, grad):
        x, weight = ctx.saved_tensors
        grad = grad.contiguous()
        dx = native().gemm_nn(grad, weight)        # [M,N] @ [N,K] -> [M,K]
        dw = native().gemm_tn(grad, x)             # [M,N]^T @ [M,K] -> [N,K]
        db = grad.sum(0) if ctx.has_bias else None
        return dx, dw, db


def linear(x: torch.Tensor, weight: torch.Tensor,
           bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    if (_use_native(x) and x.dtype == torch.bfloat16 and x.dim() == 2
            and x.shape[1] % 16 == 0 and weight.shape[0] % 16 == 0):
        return _LinearMFMA.apply(x, weight, bias)
    return F.linear(x, weight, bias)


# ---------------------------------------------------------------------------
# fused attention (flash-style forward kernel; attention.hip)
# ---------------------------------------------------------------------------

class _AttentionMFMA(torch.autograd.Function):
    """Fused forward (scores never materialised); the backward recomputes
    the probabilities from q,k (standard flash recompute composition) and
    applies the exact attention gradient formulas with batched GEMMs.
    Memory: O(S^2) only inside backward, transiently.
    """

    @staticmethod
    def forward(ctx, q, k, v, scale):
        o, lse = native().attn_fwd(q, k, v, scale)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, grad):
        q, k, v, o, lse = ctx.saved_tensors
        scale = ctx.scale
        grad = grad.contiguous()
        import os
        if os.environ.get("DLX_ATTN_BWD") == "compose":
            # recompute-composition fallback (bf16 GEMMs, fp32 softmax)
            s = (q @ k.transpose(-1, -2)).float() * scale
            p = torch.exp(s - lse.unsqueeze(-1))
            dv = p.to(q.dtype).transpose(-1, -2) @ grad
            dp = (grad @ v.transpose(-1, -2)).float()
            delta = (grad.float() * o.float()).sum(-1, keepdim=True)
            ds = (p * (dp - delta) * scale).to(q.dtype)
            return ds @ k, ds.transpose(-1, -2) @ q, dv, None
        dq, dk, dv = native().attn_bwd(grad, q, k, v, o, lse, scale)
        return dq, dk, dv, None


def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
              scale: Optional[float] = None) -> torch.Tensor:
    """q,k,v: [B,H,S,Dh].  Fused CDNA4 kernel when Dh==64, S%64==0, bf16
    on GPU (DLX_ATTN=0 forces the bmm+softmax composition)."""
    import os
    if scale is None:
        scale = q.shape[-1] ** -0.5
    if (_use_native(q) and q.dtype == torch.bfloat16 and q.shape[-1] == 64
            and q.shape[-2] % 64 == 0 and os.environ.get("DLX_ATTN") != "0"):
        return _AttentionMFMA.apply(q, k, v, scale)
    # composition fallback: explicit batched GEMM + fp32 softmax
    s = (q @ k.transpose(-1, -2)) * scale
    p = torch.softmax(s.float(), dim=-1).to(q.dtype)
    return p @ v
