"""BERT-base masked-LM on the HIP op layer (BASELINE.json config 3:
"BERT-base masked-LM fine-tune, 8 clients, FedAvg E=5").

Every hot op is a gfx950 kernel: QKV/projection/FFN Linears are MFMA GEMMs
(fused fp32-bias epilogue), attention is batched MFMA GEMMs around the
fused-scale softmax kernel (ops/functional.attention), LayerNorm and GELU
are the norm/elementwise kernels, the MLM loss is the fused log-softmax CE
over gathered masked positions, and the optimizer is the fused Adam.
Token/position embeddings are torch gathers (index ops, not in the
hand-written hot-op list).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from baton_amd.ops import functional as BF
from baton_amd.ops.modules import BatonGELU, BatonLayerNorm, BatonLinear
from baton_amd.runtime.local import LocalTrainer
from baton_amd.utils.config import TrainConfig


@dataclass
class BertConfig:
    # vocab padded to a multiple of 64 (true BERT vocab is 30522) so the
    # decoder GEMM rows stay 16-B aligned for glds staging — standard
    # tensor-core practice; labels never reference the 6 pad ids.
    vocab_size: int = 30528
    hidden: int = 768
    layers: int = 12
    heads: int = 12
    ffn: int = 3072
    max_positions: int = 512
    layer_norm_eps: float = 1e-12


class BertSelfAttention(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.heads = cfg.heads
        self.head_dim = cfg.hidden // cfg.heads
        self.qkv = BatonLinear(cfg.hidden, 3 * cfg.hidden)
        self.proj = BatonLinear(cfg.hidden, cfg.hidden)

    def forward(self, x):
        B, S, H = x.shape
        # packed QKV consumed as strided views — no permute copies on
        # either pass (AttnPackedFn writes dqkv slices in place)
        qkv = self.qkv(x).view(B, S, 3, self.heads, self.head_dim)
        o = BF.attention_qkv(qkv, causal=False)              # [B, S, h, dh]
        return self.proj(o.reshape(B, S, H))


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.attn = BertSelfAttention(cfg)
        self.ln1 = BatonLayerNorm(cfg.hidden, eps=cfg.layer_norm_eps)
        self.fc1 = BatonLinear(cfg.hidden, cfg.ffn)
        self.act = BatonGELU()
        self.fc2 = BatonLinear(cfg.ffn, cfg.hidden)
        self.ln2 = BatonLayerNorm(cfg.hidden, eps=cfg.layer_norm_eps)

    def forward(self, x):
        # post-LN, BERT style; residual adds fused into the LN kernels
        # (layer_norm_add: one pass writes z = x + sub(x) and normalizes it)
        x = BF.layer_norm_add(self.attn(x), x, self.ln1.weight,
                              self.ln1.bias, self.ln1.eps)
        x = BF.layer_norm_add(self.fc2(self.act(self.fc1(x))), x,
                              self.ln2.weight, self.ln2.bias, self.ln2.eps)
        return x


class BertForMaskedLM(nn.Module):
    """Encoder + MLM head. ``forward(input_ids)`` returns hidden states;
    ``mlm_loss`` gathers the masked positions and applies the fused CE."""

    name = "bert-base-mlm"

    def __init__(self, cfg: Optional[BertConfig] = None,
                 train_config: Optional[TrainConfig] = None):
        super().__init__()
        self.cfg = cfg = cfg or BertConfig()
        self.tok_emb = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.pos_emb = nn.Embedding(cfg.max_positions, cfg.hidden)
        self.emb_ln = BatonLayerNorm(cfg.hidden, eps=cfg.layer_norm_eps)
        self.encoder = nn.ModuleList(BertLayer(cfg) for _ in range(cfg.layers))
        self.head_fc = BatonLinear(cfg.hidden, cfg.hidden)
        self.head_act = BatonGELU()
        self.head_ln = BatonLayerNorm(cfg.hidden, eps=cfg.layer_norm_eps)
        self.decoder = BatonLinear(cfg.hidden, cfg.vocab_size)
        tc = train_config or TrainConfig(optimizer="adam", lr=5e-5)
        self._trainer = LocalTrainer(tc, loss_fn=self.mlm_loss)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device)
        x = self.tok_emb(input_ids) + self.pos_emb(pos)[None, :, :]
        x = self.emb_ln(x)
        for layer in self.encoder:
            x = layer(x)
        return x

    def mlm_logits(self, hidden: torch.Tensor, mask_positions: torch.Tensor):
        """Gather masked positions ([Nmask] flat indices into [B*S]) and
        decode to vocab logits — only masked tokens pay the vocab GEMM."""
        H = hidden.shape[-1]
        flat = hidden.reshape(-1, H)
        sel = flat.index_select(0, mask_positions)
        sel = self.head_ln(self.head_act(self.head_fc(sel)))
        return self.decoder(sel)

    def mlm_loss(self, hidden: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
        """labels: [B, S] int64, -100 on unmasked positions (HF convention).
        Only the masked rows go through the vocab GEMM + CE."""
        flat = labels.reshape(-1)
        mask_positions = (flat != -100).nonzero(as_tuple=True)[0]
        logits = self.mlm_logits(hidden, mask_positions)
        return BF.cross_entropy(logits.contiguous(), flat[mask_positions])

    def train_round(self, *data, n_epoch: int = 1) -> List[float]:
        return self._trainer(self, data, n_epoch)


def bert_base(train_config: Optional[TrainConfig] = None) -> BertForMaskedLM:
    return BertForMaskedLM(BertConfig(), train_config)


def bert_tiny(train_config: Optional[TrainConfig] = None) -> BertForMaskedLM:
    """Small config for CPU tests / quick GPU checks."""
    cfg = BertConfig(vocab_size=512, hidden=64, layers=2, heads=4, ffn=128,
                     max_positions=64)
    m = BertForMaskedLM(cfg, train_config)
    m.name = "bert-tiny-mlm"
    return m


def make_synthetic_mlm(
    n_samples: int, seq_len: int, vocab_size: int = 30522,
    mask_frac: float = 0.15, seed: int = 0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Synthetic MLM data (no network for real corpora): random token ids;
    ~15% of positions masked (token replaced by id 0 = [MASK]); labels [n,S]
    hold the original token there and -100 elsewhere (HF convention)."""
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(1, vocab_size, (n_samples, seq_len), generator=g)
    mask = torch.rand(n_samples, seq_len, generator=g) < mask_frac
    mask[:, 0] |= ~mask.any(dim=1)   # every sample has >= 1 masked position
    labels = torch.full_like(ids, -100)
    labels[mask] = ids[mask]
    masked_ids = ids.masked_fill(mask, 0)
    return masked_ids, labels
