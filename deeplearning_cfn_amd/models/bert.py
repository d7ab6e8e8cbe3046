"""BERT-base pretraining model (BASELINE.json config #4).

The reference era had no transformers; BASELINE.json adds BERT-base as
the transformer DP config ("transformer path, MFMA GEMM").  Built from
scratch on the same ops seam: Linear routes through the MFMA GEMM kernel,
LayerNorm/GELU through their HIP kernels when native is selected,
attention through torch SDPA (flash-style, memory-bound-safe in 288 GB).
Data-parallel only, per SURVEY.md §3.3 scope ruling.
"""

from __future__ import annotations

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclass
class BertConfig:
    vocab_size: int = 30522
    hidden: int = 768
    layers: int = 12
    heads: int = 12
    intermediate: int = 3072
    max_seq: int = 512
    type_vocab: int = 2
    dropout: float = 0.0          # pretraining-benchmark default: off
    layer_norm_eps: float = 1e-12


class MFMALinear(nn.Linear):
    """Linear that dispatches to the MFMA GEMM kernel on bf16/GPU."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from ..ops import functional as F_amd
        shape = x.shape
        y = F_amd.linear(x.reshape(-1, shape[-1]), self.weight, self.bias)
        return y.reshape(*shape[:-1], self.out_features)


class FusedLayerNorm(nn.LayerNorm):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from ..ops import functional as F_amd
        if hasattr(F_amd, "layer_norm"):
            return F_amd.layer_norm(x, self.normalized_shape, self.weight,
                                    self.bias, self.eps)
        return super().forward(x)


class SelfAttention(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.heads = cfg.heads
        self.dh = cfg.hidden // cfg.heads
        self.qkv = MFMALinear(cfg.hidden, 3 * cfg.hidden)
        self.proj = MFMALinear(cfg.hidden, cfg.hidden)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, S, H = x.shape
        qkv = self.qkv(x).view(B, S, 3, self.heads, self.dh)
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))
        if q.is_cuda:
            # Fused flash-style CDNA4 kernel (attention.hip) when shapes
            # qualify; otherwise explicit batched-GEMM + fp32 softmax —
            # never Triton / multi-backend dispatch.
            from ..ops import functional as F_amd
            o = F_amd.attention(q.contiguous(), k.contiguous(),
                                v.contiguous(),
                                scale=1.0 / math.sqrt(self.dh))
        else:
            o = F.scaled_dot_product_attention(q, k, v)
        return self.proj(o.transpose(1, 2).reshape(B, S, H))


class Block(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.attn = SelfAttention(cfg)
        self.ln1 = FusedLayerNorm(cfg.hidden, eps=cfg.layer_norm_eps)
        self.fc1 = MFMALinear(cfg.hidden, cfg.intermediate)
        self.fc2 = MFMALinear(cfg.intermediate, cfg.hidden)
        self.ln2 = FusedLayerNorm(cfg.hidden, eps=cfg.layer_norm_eps)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from ..ops import functional as F_amd
        x = self.ln1(x + self.attn(x))
        h = self.fc1(x)
        h = F_amd.gelu(h) if hasattr(F_amd, "gelu") else F.gelu(h)
        return self.ln2(x + self.fc2(h))


class BertForPretraining(nn.Module):
    """Encoder + MLM head (+NSP head) — the classic pretrain benchmark."""

    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        # vocab padded to a multiple of 64 so the weight-tied MLM-head
        # GEMM meets the MFMA kernel's alignment gate;
        # pad rows are never labeled, CE simply learns them away
        self.vocab_padded = (cfg.vocab_size + 63) // 64 * 64
        self.tok = nn.Embedding(self.vocab_padded, cfg.hidden)
        self.pos = nn.Embedding(cfg.max_seq, cfg.hidden)
        self.typ = nn.Embedding(cfg.type_vocab, cfg.hidden)
        self.ln = FusedLayerNorm(cfg.hidden, eps=cfg.layer_norm_eps)
        self.blocks = nn.ModuleList(Block(cfg) for _ in range(cfg.layers))
        self.mlm_dense = MFMALinear(cfg.hidden, cfg.hidden)
        self.mlm_ln = FusedLayerNorm(cfg.hidden, eps=cfg.layer_norm_eps)
        self.nsp = MFMALinear(cfg.hidden, 2)
        self.apply(self._init)

    @staticmethod
    def _init(m: nn.Module) -> None:
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def encode(self, input_ids: torch.Tensor,
               token_type: torch.Tensor | None = None) -> torch.Tensor:
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device).unsqueeze(0)
        typ = token_type if token_type is not None else torch.zeros_like(input_ids)
        x = self.ln(self.tok(input_ids) + self.pos(pos) + self.typ(typ))
        for blk in self.blocks:
            x = blk(x)
        return x

    def forward(self, input_ids: torch.Tensor,
                mlm_labels: torch.Tensor | None = None,
                nsp_labels: torch.Tensor | None = None):
        x = self.encode(input_ids)
        h = self.mlm_ln(F.gelu(self.mlm_dense(x)))
        # MLM logits via weight tying with the token embedding; routed
        # through the MFMA GEMM (padded vocab keeps N % 16 == 0)
        from ..ops import functional as F_amd
        B, S, H = h.shape
        logits = F_amd.linear(h.reshape(-1, H), self.tok.weight)
        logits = logits.reshape(B, S, self.vocab_padded)
        if mlm_labels is None:
            return logits
        loss = F.cross_entropy(
            logits.float().view(-1, self.vocab_padded),
            mlm_labels.view(-1), ignore_index=-100)
        if nsp_labels is not None:
            cls = self.nsp(x[:, 0])
            loss = loss + F.cross_entropy(cls.float(), nsp_labels)
        return loss


def build_bert(size: str = "base", seq_len: int = 128) -> BertForPretraining:
    if size == "base":
        cfg = BertConfig()
    elif size == "large":
        cfg = BertConfig(hidden=1024, layers=24, heads=16, intermediate=4096)
    else:
        raise ValueError(f"unknown bert size {size!r}")
    cfg.max_seq = max(cfg.max_seq, seq_len)
    return BertForPretraining(cfg)
