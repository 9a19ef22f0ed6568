"""HIP/CDNA4 kernel layer.

The reference has zero native code; its compute runs on
torch-CPU ops. Here every hot op of the federated training loop is a
hand-written gfx950 HIP kernel (csrc/), exposed through torch.autograd
wrappers:

  ops.linear      — MFMA GEMM forward/backward (Linear layers)
  ops.conv        — implicit-GEMM convolution (ResNet configs)
  ops.norm        — LayerNorm / BatchNorm forward/backward
  ops.optim       — fused SGD / fused Adam (single kernel over flat buffers)
  ops.fedmath     — axpby / scale kernels for the FedAvg pre-scale
  ops.loss        — MSE / cross-entropy forward/backward

On a machine without a GPU the wrappers fall back to stock torch ops so the
CPU test-suite runs everywhere; on a GPU box a missing extension raises —
the HIP path is the product, not an option (see ops/_ext.py).
"""

from baton_amd.ops._ext import hip_ops, hip_available, require_hip

__all__ = ["hip_ops", "hip_available", "require_hip"]
