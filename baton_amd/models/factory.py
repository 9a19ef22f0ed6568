"""Model + synthetic-data factory shared by the launcher, bench and tests.

Names cover the BASELINE.json configs:
  linreg10      — reference demo model (config 1 oracle)
  tinymlp       — CPU plumbing config
  resnet18/50   — headline federated configs (2, 5)
  bert-base / bert-tiny   — masked-LM config (3)
  llama-lora / llama-tiny — LoRA fine-tune config (4)
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from baton_amd.utils.config import TrainConfig


def create_model(name: str, train_config: Optional[TrainConfig] = None):
    if name == "linreg10":
        from baton_amd.models.mlp import LinearRegressionModel

        return LinearRegressionModel(train_config)
    if name == "tinymlp":
        from baton_amd.models.mlp import TinyMLP

        return TinyMLP(train_config=train_config)
    if name in ("resnet18", "resnet50"):
        from baton_amd.models.resnet import resnet18, resnet50

        return (resnet18 if name == "resnet18" else resnet50)(
            train_config=train_config
        )
    if name in ("bert-base", "bert-tiny"):
        from baton_amd.models.bert import bert_base, bert_tiny

        return (bert_base if name == "bert-base" else bert_tiny)(train_config)
    if name in ("llama-lora", "llama-tiny"):
        from baton_amd.models.llama import (
            LlamaForCausalLM,
            llama3_8b_config,
            llama_tiny_config,
        )

        cfg = llama3_8b_config() if name == "llama-lora" else llama_tiny_config()
        return LlamaForCausalLM(cfg, train_config)
    raise ValueError(f"unknown model {name!r}")


def make_data(name: str, n_samples: int, seed: int = 0, seq_len: int = 128,
              dtype: torch.dtype = torch.float32) -> Tuple[tuple, int]:
    """Synthetic data shaped for ``name``; returns ((inputs..., target), n)."""
    if name in ("linreg10", "tinymlp"):
        from baton_amd.models.mlp import make_synthetic_regression

        x, y = make_synthetic_regression(n_samples, seed=seed)
        return (x.to(dtype), y.to(dtype)), n_samples
    if name in ("resnet18", "resnet50"):
        from baton_amd.models.resnet import make_synthetic_cifar

        x, y = make_synthetic_cifar(n_samples, seed=seed, dtype=dtype)
        return (x, y), n_samples
    if name in ("bert-base", "bert-tiny"):
        from baton_amd.models.bert import make_synthetic_mlm

        vocab = 30522 if name == "bert-base" else 512  # data ids < true vocab
        if name == "bert-tiny":
            seq_len = min(seq_len, 64)   # tiny config max_positions
        ids, labels = make_synthetic_mlm(n_samples, seq_len, vocab, seed=seed)
        return (ids, labels), n_samples
    if name in ("llama-lora", "llama-tiny"):
        from baton_amd.models.llama import make_synthetic_clm

        vocab = 128256 if name == "llama-lora" else 512
        if name == "llama-tiny":
            seq_len = min(seq_len, 128)  # tiny config max_positions
        ids, labels = make_synthetic_clm(n_samples, seq_len, vocab, seed=seed)
        return (ids, labels), n_samples
    raise ValueError(f"unknown model {name!r}")
