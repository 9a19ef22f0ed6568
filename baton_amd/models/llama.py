"""Llama-3-style causal LM with LoRA adapters, on the HIP op layer
(BASELINE.json config 4: "Llama-3 8B LoRA federated fine-tune — only LoRA
deltas reduced over xGMI").

Architecture (Llama-3 shapes): RMSNorm -> GQA attention with RoPE ->
RMSNorm -> SwiGLU MLP; weights bf16, random-init (no network for real
checkpoints). Base weights are FROZEN; only the LoRA A/B adapters train —
LinearFn skips the base wgrad GEMM entirely (needs_input_grad), and the
federated round reduces only the adapter deltas (a few MB, latency-bound)
instead of the 16 GB base model.

Hot ops: MFMA GEMMs (all projections + LoRA), batched-GEMM causal
attention with the fused-scale softmax kernel, RMSNorm / RoPE / SwiGLU
kernels, fused CE, fused Adam on the adapter arena.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from baton_amd.ops import functional as BF
from baton_amd.ops.modules import BatonLinear
from baton_amd.runtime.local import LocalTrainer
from baton_amd.utils.config import TrainConfig


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden: int = 4096
    layers: int = 32
    heads: int = 32
    kv_heads: int = 8
    ffn: int = 14336
    max_positions: int = 8192
    rope_base: float = 500000.0
    rms_eps: float = 1e-5
    lora_rank: int = 16
    lora_alpha: float = 32.0


def llama3_8b_config() -> LlamaConfig:
    return LlamaConfig()


def llama_tiny_config() -> LlamaConfig:
    return LlamaConfig(vocab_size=512, hidden=128, layers=2, heads=4,
                       kv_heads=2, ffn=256, max_positions=128, lora_rank=4)


class LoRALinear(nn.Module):
    """y = x @ W^T + (alpha/r) * (x @ A^T) @ B^T; W frozen, A/B trainable.
    A: [r, in] (kaiming init), B: [out, r] (zero init) — standard LoRA."""

    def __init__(self, in_features: int, out_features: int, rank: int,
                 alpha: float):
        super().__init__()
        self.weight = nn.Parameter(
            torch.empty(out_features, in_features), requires_grad=False
        )
        nn.init.normal_(self.weight, 0.0, 0.02)
        self.lora_a = nn.Parameter(torch.empty(rank, in_features))
        self.lora_b = nn.Parameter(torch.zeros(out_features, rank))
        nn.init.kaiming_uniform_(self.lora_a, a=math.sqrt(5))
        self.scaling = alpha / rank
        self._weight_t = None   # frozen-weight transpose, cached per device
        self._weight_t_version = -1

    def _wt(self):
        # the base weight is FROZEN: cache W^T once so every backward's
        # dgrad runs on the glds NT path without re-transposing. Keyed on
        # device/dtype AND the tensor's in-place version counter, so a
        # load_state_dict into an already-run model (which mutates the
        # frozen weight in place) invalidates the cache.
        if (self._weight_t is None
                or self._weight_t.device != self.weight.device
                or self._weight_t.dtype != self.weight.dtype
                or self._weight_t_version != self.weight._version):
            with torch.no_grad():
                self._weight_t = self.weight.t().contiguous()
            self._weight_t_version = self.weight._version
        return self._weight_t

    def forward(self, x):
        # cached W^T serves the in-house NT dgrad only; on the library
        # path it would just double the frozen-weight memory (~3 GB at 8B)
        wt = (self._wt() if (x.is_cuda and not self.weight.requires_grad
                             and not BF.lib_gemm_enabled()) else None)
        # combine fused into the rank-r B GEMM's C-accumulate epilogue
        return BF.lora_linear(x, self.weight, self.lora_a, self.lora_b,
                              self.scaling, weight_t=wt)


class LlamaAttention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.head_dim = cfg.hidden // cfg.heads
        kv_dim = cfg.kv_heads * self.head_dim
        self.q_proj = LoRALinear(cfg.hidden, cfg.hidden, cfg.lora_rank, cfg.lora_alpha)
        self.k_proj = LoRALinear(cfg.hidden, kv_dim, cfg.lora_rank, cfg.lora_alpha)
        self.v_proj = LoRALinear(cfg.hidden, kv_dim, cfg.lora_rank, cfg.lora_alpha)
        self.o_proj = LoRALinear(cfg.hidden, cfg.hidden, cfg.lora_rank, cfg.lora_alpha)

    def forward(self, x, cos_t, sin_t):
        B, S, H = x.shape
        cfg = self.cfg
        dh = self.head_dim
        q = self.q_proj(x).reshape(B, S, cfg.heads, dh)
        k = self.k_proj(x).reshape(B, S, cfg.kv_heads, dh)
        v = self.v_proj(x).reshape(B, S, cfg.kv_heads, dh)
        q = BF.rope(q, cos_t, sin_t)
        k = BF.rope(k, cos_t, sin_t)
        # GQA: KV heads stay un-replicated AND un-permuted — the strided
        # batched GEMM consumes [B,S,h,dh] views in place (b_group indexes
        # the shared KV head in-kernel; no repeat_interleave, no permutes)
        o = BF.attention_bshd(q, k, v, causal=True)          # [B, S, h, dh]
        return self.o_proj(o.reshape(B, S, H))


class LlamaMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.gate = LoRALinear(cfg.hidden, cfg.ffn, cfg.lora_rank, cfg.lora_alpha)
        self.up = LoRALinear(cfg.hidden, cfg.ffn, cfg.lora_rank, cfg.lora_alpha)
        self.down = LoRALinear(cfg.ffn, cfg.hidden, cfg.lora_rank, cfg.lora_alpha)

    def forward(self, x):
        return self.down(BF.silu_mul(self.gate(x), self.up(x)))


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim), requires_grad=False)
        self.eps = eps

    def forward(self, x):
        return BF.rms_norm(x, self.weight, self.eps)


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.attn_norm = RMSNorm(cfg.hidden, cfg.rms_eps)
        self.attn = LlamaAttention(cfg)
        self.mlp_norm = RMSNorm(cfg.hidden, cfg.rms_eps)
        self.mlp = LlamaMLP(cfg)

    def forward(self, x, pend, cos_t, sin_t):
        """Pending-pair form: the true residual stream is x (+ pend).
        Each norm consumes the pending add via the fused add_rms_norm
        kernel (one pass writes z = x + pend and normalizes it), so NO
        standalone residual-add kernel runs anywhere in the model —
        including across layer boundaries. Returns (z, pending')."""
        if pend is None:
            n1, z1 = self.attn_norm(x), x
        else:
            n1, z1 = BF.add_rms_norm(x, pend, self.attn_norm.weight,
                                     self.attn_norm.eps)
        a = self.attn(n1, cos_t, sin_t)
        n2, z2 = BF.add_rms_norm(z1, a, self.mlp_norm.weight,
                                 self.mlp_norm.eps)
        return z2, self.mlp(n2)


class LlamaForCausalLM(nn.Module):
    """Causal LM fine-tune: next-token CE. Embeddings + lm_head frozen
    (the LoRA setup trains adapters only)."""

    name = "llama-lora"

    def __init__(self, cfg: LlamaConfig, train_config: Optional[TrainConfig] = None):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.embed.weight.requires_grad_(False)
        self.layers = nn.ModuleList(LlamaDecoderLayer(cfg) for _ in range(cfg.layers))
        self.final_norm = RMSNorm(cfg.hidden, cfg.rms_eps)
        self.lm_head = BatonLinear(cfg.hidden, cfg.vocab_size, bias=False)
        self.lm_head.weight.requires_grad_(False)
        head_dim = cfg.hidden // cfg.heads
        cos, sin = BF.rope_tables(cfg.max_positions, head_dim, cfg.rope_base)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        tc = train_config or TrainConfig(optimizer="adam", lr=1e-4)
        self._trainer = LocalTrainer(tc, loss_fn=self.lm_loss)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        S = input_ids.shape[1]
        x = self.embed(input_ids)
        cos = self.rope_cos[:S].contiguous()
        sin = self.rope_sin[:S].contiguous()
        pend = None
        for layer in self.layers:
            x, pend = layer(x, pend, cos, sin)
        if pend is None:
            return self.final_norm(x)
        y, _ = BF.add_rms_norm(x, pend, self.final_norm.weight,
                               self.final_norm.eps)
        return y

    def lm_loss(self, hidden: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
        """Next-token CE: hidden [B,S,H], labels [B,S] (input shifted by
        caller or identical ids — we shift internally)."""
        B, S, H = hidden.shape
        logits = self.lm_head(hidden[:, :-1, :].reshape(-1, H).contiguous())
        targets = labels[:, 1:].reshape(-1).contiguous()
        return BF.cross_entropy(logits.contiguous(), targets)

    def train_round(self, *data, n_epoch: int = 1) -> List[float]:
        return self._trainer(self, data, n_epoch)

    def lora_parameters(self):
        return [p for p in self.parameters() if p.requires_grad]

    def lora_state_dict(self):
        """Only adapter tensors — the federated payload (adapter-delta-only
        reduce; the 16 GB base never crosses xGMI)."""
        from collections import OrderedDict

        return OrderedDict(
            (k, v) for k, v in self.state_dict().items() if "lora_" in k
        )


def llama_lora(cfg: Optional[LlamaConfig] = None,
               train_config: Optional[TrainConfig] = None) -> LlamaForCausalLM:
    return LlamaForCausalLM(cfg or llama3_8b_config(), train_config)


def make_synthetic_clm(n_samples: int, seq_len: int, vocab_size: int,
                       seed: int = 0) -> Tuple[torch.Tensor, torch.Tensor]:
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, vocab_size, (n_samples, seq_len), generator=g)
    return ids, ids.clone()
