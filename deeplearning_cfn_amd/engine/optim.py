"""Optimizers over the engine's flat buckets.

The reference's server-side SGD aggregation (PS path) and per-rank
``apply_gradients`` (Horovod path) collapse to: every rank applies the
identical update to the identical averaged gradients.

``FlatSGD`` walks the DataParallel engine's buckets and updates each flat
param/grad pair with ONE fused HIP kernel launch per bucket
(sgd_momentum_flat: wd + momentum + nesterov + lr in a single pass,
fp32 momentum for bf16 params).  Torch fallback keeps CPU tests running.
``FlatAdam`` is the BERT-path equivalent (fused adamw_flat kernel).
"""

from __future__ import annotations

from typing import List, Optional

import torch

from ..ops import native, native_available
from .ddp import DataParallel


class _FlatOptimBase:
    def __init__(self, engine: DataParallel):
        self.engine = engine
        self.buckets = engine.plan.buckets
        # one contiguous param mirror per bucket so the fused kernel walks
        # a single pointer; params are re-viewed into it (zero-copy update).
        self.flat_params: List[torch.Tensor] = []
        # fp32 master weights for reduced-precision params (bf16 training):
        # the fused kernels update the master and write the bf16 copy.
        self.masters: List[Optional[torch.Tensor]] = []
        for b in self.buckets:
            flat_p = torch.empty_like(b.flat)
            off = 0
            with torch.no_grad():
                for p in b.params:
                    flat_p[off:off + p.numel()].copy_(p.detach().reshape(-1))
                    # re-point the parameter at its slice of the flat buffer
                    p.data = flat_p[off:off + p.numel()].view_as(p)
                    off += p.numel()
            self.flat_params.append(flat_p)
            self.masters.append(flat_p.float()
                                if flat_p.dtype != torch.float32 else None)

    def zero_grad(self) -> None:
        self.engine.zero_grad()

    def refresh_masters(self) -> None:
        """Re-derive fp32 master weights from the (bf16) params.

        Needed after ``model.load_state_dict`` — loading copies into the
        flat param buffers, leaving masters captured at construction time
        stale.  Checkpoints that carry optimizer state overwrite these
        again in ``load_state_dict`` (SURVEY.md §6.4 resume contract).
        """
        for fp, master in zip(self.flat_params, self.masters):
            if master is not None:
                master.copy_(fp.float())

    def state_dict(self) -> dict:
        raise NotImplementedError

    def load_state_dict(self, sd: dict) -> None:
        raise NotImplementedError


class FlatSGD(_FlatOptimBase):
    def __init__(self, engine: DataParallel, lr: float = 0.1,
                 momentum: float = 0.9, weight_decay: float = 1e-4,
                 nesterov: bool = False):
        super().__init__(engine)
        self.lr, self.momentum = lr, momentum
        self.weight_decay, self.nesterov = weight_decay, nesterov
        self.momentum_bufs = [
            torch.zeros(fp.numel(), dtype=torch.float32, device=fp.device)
            for fp in self.flat_params]

    @torch.no_grad()
    def step(self) -> None:
        for fp, b, mbuf, master in zip(self.flat_params, self.buckets,
                                       self.momentum_bufs, self.masters):
            g = b.flat
            if fp.is_cuda and native_available():
                if master is not None:
                    native().sgd_momentum_flat_master(
                        fp, g, mbuf, master, self.lr, self.momentum,
                        self.weight_decay, self.nesterov)
                else:
                    native().sgd_momentum_flat(
                        fp, g, mbuf, self.lr, self.momentum,
                        self.weight_decay, self.nesterov)
                continue
            # torch fallback — numerically the reference update:
            # m = mu*m + (g + wd*p); p -= lr * (m or g+mu*m)
            pv = master if master is not None else fp
            g32 = g.float()
            if self.weight_decay:
                g32 = g32.add(pv.float(), alpha=self.weight_decay)
            mbuf.mul_(self.momentum).add_(g32)
            upd = g32.add(mbuf, alpha=self.momentum) if self.nesterov else mbuf
            if master is not None:
                master.sub_(self.lr * upd)
                fp.copy_(master.to(fp.dtype))
            else:
                fp.copy_((fp.float() - self.lr * upd).to(fp.dtype))

    def state_dict(self) -> dict:
        return {"momentum_bufs": self.momentum_bufs, "masters": self.masters,
                "lr": self.lr, "momentum": self.momentum,
                "weight_decay": self.weight_decay, "nesterov": self.nesterov}

    def load_state_dict(self, sd: dict) -> None:
        for dst, src in zip(self.momentum_bufs, sd["momentum_bufs"]):
            dst.copy_(src.to(dst.device))
        for dst, src in zip(self.masters, sd.get("masters", [])):
            if dst is not None and src is not None:
                dst.copy_(src.to(dst.device))
        self.lr = sd["lr"]
        self.momentum = sd["momentum"]
        self.weight_decay = sd["weight_decay"]
        self.nesterov = sd["nesterov"]


class FlatAdam(_FlatOptimBase):
    """AdamW over flat buckets (BERT path; fused adamw_flat on HIP)."""

    def __init__(self, engine: DataParallel, lr: float = 1e-4,
                 betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.01):
        super().__init__(engine)
        self.lr, self.betas, self.eps = lr, betas, eps
        self.weight_decay = weight_decay
        self.t = 0
        self.m = [torch.zeros(fp.numel(), dtype=torch.float32,
                              device=fp.device) for fp in self.flat_params]
        self.v = [torch.zeros_like(mm) for mm in self.m]

    @torch.no_grad()
    def step(self) -> None:
        self.t += 1
        b1, b2 = self.betas
        bc1 = 1 - b1 ** self.t
        bc2 = 1 - b2 ** self.t
        for fp, b, m, v, master in zip(self.flat_params, self.buckets,
                                       self.m, self.v, self.masters):
            g = b.flat
            if fp.is_cuda and native_available():
                if master is not None:
                    native().adamw_flat_master(fp, g, m, v, master, self.lr,
                                               b1, b2, self.eps,
                                               self.weight_decay, bc1, bc2)
                else:
                    native().adamw_flat(fp, g, m, v, self.lr, b1, b2,
                                        self.eps, self.weight_decay, bc1, bc2)
                continue
            g32 = g.float()
            m.mul_(b1).add_(g32, alpha=1 - b1)
            v.mul_(b2).addcmul_(g32, g32, value=1 - b2)
            p32 = master if master is not None else fp.float()
            p32 = p32 - self.lr * self.weight_decay * p32
            p32 = p32 - self.lr * (m / bc1) / ((v / bc2).sqrt() + self.eps)
            if master is not None:
                master.copy_(p32)
            fp.copy_(p32.to(fp.dtype))

    def state_dict(self) -> dict:
        return {"m": self.m, "v": self.v, "masters": self.masters,
                "t": self.t, "lr": self.lr, "betas": self.betas,
                "eps": self.eps, "weight_decay": self.weight_decay}

    def load_state_dict(self, sd: dict) -> None:
        for dst, src in zip(self.m, sd["m"]):
            dst.copy_(src.to(dst.device))
        for dst, src in zip(self.v, sd["v"]):
            dst.copy_(src.to(dst.device))
        for dst, src in zip(self.masters, sd.get("masters", [])):
            if dst is not None and src is not None:
                dst.copy_(src.to(dst.device))
        self.t = sd["t"]
        self.lr = sd["lr"]
        self.betas = tuple(sd.get("betas", self.betas))
        self.eps = sd.get("eps", self.eps)
        self.weight_decay = sd.get("weight_decay", self.weight_decay)
