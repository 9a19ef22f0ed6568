"""nn.Module wrappers over the HIP op layer (ops/functional.py).

These are the building blocks of the model zoo (models/): drop-in-shaped
replacements for nn.Linear / nn.Conv2d / nn.LayerNorm / nn.BatchNorm2d that
route through the gfx950 kernels on GPU. Conventions:

  * activations NHWC ([N,H,W,C]) — channels contiguous is the natural
    vector-load layout for CDNA4 (16 B/lane along C);
  * conv weights [Cout, KH, KW, Cin];
  * BatchNorm params/stats fp32 regardless of activation dtype (classic
    mixed-precision practice; FedAvg aggregates them exactly);
  * Linear/LayerNorm params follow the model dtype (bf16 configs keep bf16
    master-less params — the fused SGD/Adam kernels handle both).
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn

from baton_amd.ops import functional as BF


class BatonLinear(nn.Module):
    def __init__(self, in_features: int, out_features: int, bias: bool = True):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        # bias kept fp32: it feeds the GEMM epilogue's fp32 bias port
        self.bias = nn.Parameter(torch.empty(out_features, dtype=torch.float32)) if bias else None
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        if self.bias is not None:
            bound = 1 / math.sqrt(self.in_features)
            nn.init.uniform_(self.bias, -bound, bound)

    def forward(self, x):
        return BF.linear(x, self.weight, self.bias)

    def _apply(self, fn, recurse=True):
        # keep bias fp32 through .to(bf16): re-cast after generic _apply
        out = super()._apply(fn, recurse)
        if out.bias is not None and out.bias.dtype != torch.float32:
            out.bias.data = out.bias.data.float()
        return out

    def extra_repr(self):
        return f"in={self.in_features}, out={self.out_features}, bias={self.bias is not None}"


class BatonConv2d(nn.Module):
    """NHWC conv, no bias (the ResNet configs follow conv->BN)."""

    def __init__(self, in_channels: int, out_channels: int, kernel_size: int,
                 stride: int = 1, padding: int = 0):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding
        self.weight = nn.Parameter(
            torch.empty(out_channels, kernel_size, kernel_size, in_channels)
        )
        self.reset_parameters()

    def reset_parameters(self):
        fan_in = self.in_channels * self.kernel_size**2
        nn.init.normal_(self.weight, 0.0, math.sqrt(2.0 / fan_in))

    def forward(self, x):
        return BF.conv2d(x, self.weight, self.stride, self.padding)

    def extra_repr(self):
        return (f"{self.in_channels}->{self.out_channels}, k={self.kernel_size}, "
                f"s={self.stride}, p={self.padding}, NHWC")


class BatonBatchNorm2d(nn.Module):
    """BatchNorm over the channel (last) dim of NHWC input, optional fused
    ReLU. Params/stats always fp32."""

    def __init__(self, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1, fused_relu: bool = False):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.fused_relu = fused_relu
        self.weight = nn.Parameter(torch.ones(num_features, dtype=torch.float32))
        self.bias = nn.Parameter(torch.zeros(num_features, dtype=torch.float32))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked", torch.tensor(0, dtype=torch.long))

    def _apply(self, fn, recurse=True):
        out = super()._apply(fn, recurse)
        # pin params/stats to fp32 across model-wide .to(bf16)
        for name in ("weight", "bias"):
            p = getattr(out, name)
            if p is not None and p.dtype != torch.float32:
                p.data = p.data.float()
        for name in ("running_mean", "running_var"):
            b = getattr(out, name)
            if b is not None and b.dtype != torch.float32:
                setattr(out, name, b.float())
        return out

    def forward(self, x):
        if self.training:
            self.num_batches_tracked += 1
            return BF.BatchNormFn.apply(
                x, self.weight, self.bias, self.running_mean, self.running_var,
                self.momentum, self.eps, self.fused_relu,
            )
        return BF.batch_norm_eval(
            x, self.weight, self.bias, self.running_mean, self.running_var,
            self.eps, self.fused_relu,
        )

    def forward_add_relu(self, x, res):
        """y = relu(bn(x) + res): the ResNet residual join fused into the
        normalize pass (training); eval falls back to eval-bn + add_relu."""
        if self.training:
            self.num_batches_tracked += 1
            return BF.BatchNormAddReLUFn.apply(
                x, res, self.weight, self.bias, self.running_mean,
                self.running_var, self.momentum, self.eps,
            )
        y = BF.batch_norm_eval(
            x, self.weight, self.bias, self.running_mean, self.running_var,
            self.eps, False,
        )
        return BF.add_relu(y, res)

    def extra_repr(self):
        return f"{self.num_features}, fused_relu={self.fused_relu}"


class BatonLayerNorm(nn.Module):
    def __init__(self, normalized_shape: int, eps: float = 1e-5):
        super().__init__()
        self.normalized_shape = normalized_shape
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(normalized_shape))
        self.bias = nn.Parameter(torch.zeros(normalized_shape))

    def forward(self, x):
        return BF.layer_norm(x, self.weight, self.bias, self.eps)


class BatonReLU(nn.Module):
    def forward(self, x):
        return BF.relu(x)


class BatonGELU(nn.Module):
    def forward(self, x):
        return BF.gelu(x)
