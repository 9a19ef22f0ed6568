"""Serialized-dispatch sanity lane (SURVEY.md §5 'race detection').

Runs a representative kernel set under ``AMD_SERIALIZE_KERNEL=3`` (every
kernel launch synchronizes and reports its own fault) in a subprocess, so a
kernel that only passes because an adjacent launch's side effects hide a
race/fault is caught. This is the ROCm analog of a compute-sanitizer CI job.
"""

import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

_SNIPPET = r"""
import torch
from baton_amd.ops import functional as BF
from baton_amd.ops._ext import require_hip

torch.manual_seed(0)
dev = "cuda"
ops = require_hip()

# GEMM fwd/bwd
a = torch.randn(256, 320, device=dev, dtype=torch.bfloat16, requires_grad=True)
w = torch.randn(192, 320, device=dev, dtype=torch.bfloat16, requires_grad=True)
y = BF.linear(a, w)
y.float().square().mean().backward()
ref = (a.detach().float() @ w.detach().float().t())
assert (y.float() - ref).abs().max() < 0.5, "gemm mismatch under serialize"

# LayerNorm fwd/bwd
x = torch.randn(128, 256, device=dev, dtype=torch.bfloat16, requires_grad=True)
g = torch.ones(256, device=dev, dtype=torch.bfloat16, requires_grad=True)
b = torch.zeros(256, device=dev, dtype=torch.bfloat16, requires_grad=True)
out = BF.layer_norm(x, g, b, 1e-5)
out.float().sum().backward()

# attention fwd/bwd (packed)
qkv = torch.randn(2, 128, 3, 4, 64, device=dev, dtype=torch.bfloat16,
                  requires_grad=True)
o = BF.attention_qkv(qkv)
o.float().square().mean().backward()

# fused optimizer
p = torch.randn(1000, device=dev, dtype=torch.float32)
gr = torch.randn_like(p)
m = torch.zeros_like(p)
ops.sgd_step(p, gr, m, 0.01, 0.9, 0.0)
torch.cuda.synchronize()
print("SERIALIZED-OK")
"""


def test_kernels_under_serialized_dispatch():
    env = dict(os.environ)
    env["AMD_SERIALIZE_KERNEL"] = "3"
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.run(
        [sys.executable, "-c", _SNIPPET],
        env=env, capture_output=True, text=True, timeout=600,
    )
    assert proc.returncode == 0, (
        f"serialized-dispatch run failed:\n{proc.stdout[-2000:]}\n{proc.stderr[-4000:]}"
    )
    assert "SERIALIZED-OK" in proc.stdout
