"""torch.autograd wrappers over the gfx950 HIP kernels.

Dispatch policy (see ops/_ext.py): on GPU the HIP kernels are mandatory —
``require_hip()`` raises if the extension is missing, never a silent eager
fallback; on CPU each Function falls back to the torch reference
implementation so the control-plane test-suite runs GPU-free. The GPU tests
(tests/test_kernels_gpu.py) compare every kernel against plain torch fp32
references (SURVEY.md §4 "kernel unit tests").
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn.functional as F

from baton_amd.ops._ext import require_hip


def _on_gpu(*tensors: torch.Tensor) -> bool:
    return tensors[0].is_cuda


# Plain (unfused) large GEMMs go to the library: hipBLASLt via torch.matmul
# measured 1455-1554 TF vs the in-house 8-phase's 815-977 at the training
# NT shapes (benchmarks/vs_blaslt.py) — a 1.5-1.6x kernel-level gap the
# schedule cannot close this round. The hand-written kernels keep every
# FUSED op (flash attention, conv, norms, optimizers, bias/residual/LoRA
# epilogues, split-K slabs) and remain the only path for them; plain
# matmuls are exactly what the library is for. BATON_LIB_GEMM=0 reverts
# to the in-house GEMM everywhere (A/B + fallback).
_LIB_GEMM = os.environ.get("BATON_LIB_GEMM", "1") != "0"


def _lib_gemm(x: torch.Tensor) -> bool:
    return _LIB_GEMM and x.dtype == torch.bfloat16


def lib_gemm_enabled() -> bool:
    """True when plain bf16 GEMMs route to the library (module callers use
    this to skip building in-house-only artifacts like cached W^T)."""
    return _LIB_GEMM


class LinearFn(torch.autograd.Function):
    """y = x @ W^T + b. x:[M,K], W:[N,K], b fp32 [N] or None.
    Forward: MFMA GEMM NT with fused bias; backward: NN dgrad + TN wgrad.
    ``weight_t`` (optional, [K,N]): a cached transpose for FROZEN weights —
    dgrad then runs directly on the glds NT path instead of re-transposing
    the weight every backward (LoRA base / lm_head)."""

    @staticmethod
    def forward(ctx, x, weight, bias, weight_t=None):
        ctx.save_for_backward(x, weight, weight_t)
        ctx.has_bias = bias is not None
        if _on_gpu(x):
            if _lib_gemm(x):
                if bias is None:
                    return x @ weight.t()
                return torch.addmm(bias.to(x.dtype), x, weight.t())
            ops = require_hip()
            return ops.gemm(x, weight, 0, bias, False, False, 1.0, 0.0)
        out = x @ weight.t()
        if bias is not None:
            out = out + bias.to(out.dtype)
        return out

    @staticmethod
    def backward(ctx, dy):
        x, weight, weight_t = ctx.saved_tensors
        dy = dy.contiguous()
        need_dx, need_dw, need_db = ctx.needs_input_grad[:3]
        dx = dw = db = None
        if _on_gpu(dy):
            ops = require_hip()
            if need_dx:
                if _lib_gemm(dy):
                    dx = dy @ weight                # library NN (native)
                elif weight_t is not None:
                    dx = ops.gemm(dy, weight_t, 0)  # NT on cached W^T [K,N]
                else:
                    dx = ops.gemm(dy, weight, 1)    # NN: dY @ W
            if need_dw:                           # skipped for frozen (LoRA base)
                if _lib_gemm(dy):
                    # library TN handles the transposed view natively —
                    # no re-layout kernels at all
                    dw = (dy.t() @ x).to(weight.dtype)
                elif dy.shape[1] <= 32:
                    # skinny-N wgrad (LoRA A): dW = dY^T @ X as a DIRECT
                    # NN — transpose only the tiny [M, r] grad; the TN
                    # route would transpose the big [M, K] activation
                    dw = ops.gemm(dy.t().contiguous(), x, 1, None, False,
                                  False, 1.0, 0.0, None, True)
                    dw = dw.to(weight.dtype)
                else:
                    dw = ops.gemm(dy, x, 2).to(weight.dtype)  # TN: dY^T @ X
            if need_db and ctx.has_bias:
                db = ops.colsum(dy)
        else:
            if need_dx:
                dx = dy @ weight
            if need_dw:
                dw = (dy.t() @ x).to(weight.dtype)
            if need_db and ctx.has_bias:
                db = dy.sum(0, dtype=torch.float32)
        return dx, dw, db, None


def linear(x: torch.Tensor, weight: torch.Tensor,
           bias: Optional[torch.Tensor] = None,
           weight_t: Optional[torch.Tensor] = None):
    shape = x.shape
    x2 = x.reshape(-1, shape[-1]).contiguous()
    y = LinearFn.apply(x2, weight, bias, weight_t)
    return y.reshape(*shape[:-1], weight.shape[0])


class Conv2dFn(torch.autograd.Function):
    """NHWC convolution: x [N,H,W,Cin], w [Cout,KH,KW,Cin] -> y [N,HO,WO,Cout].
    Implicit-GEMM HIP kernels fwd/dgrad/wgrad (conv.hip)."""

    @staticmethod
    def forward(ctx, x, w, stride: int, pad: int):
        ctx.save_for_backward(x, w)
        ctx.stride, ctx.pad = stride, pad
        if _on_gpu(x):
            ops = require_hip()
            return ops.conv_fwd(x, w, stride, pad)
        # CPU reference: NCHW conv with permutes
        xn = x.permute(0, 3, 1, 2)
        wn = w.permute(0, 3, 1, 2)
        y = F.conv2d(xn.float(), wn.float(), stride=stride, padding=pad)
        return y.permute(0, 2, 3, 1).contiguous().to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        stride, pad = ctx.stride, ctx.pad
        dy = dy.contiguous()
        if _on_gpu(dy):
            ops = require_hip()
            dx = ops.conv_dgrad(dy, w, x.shape[1], x.shape[2], stride, pad)
            dw = ops.conv_wgrad(dy, x, w.shape[1], w.shape[2], stride, pad, False)
        else:
            xn = x.permute(0, 3, 1, 2).float()
            wn = w.permute(0, 3, 1, 2).float()
            dyn = dy.permute(0, 3, 1, 2).float()
            dxn = torch.nn.grad.conv2d_input(xn.shape, wn, dyn, stride=stride, padding=pad)
            dwn = torch.nn.grad.conv2d_weight(xn, wn.shape, dyn, stride=stride, padding=pad)
            dx = dxn.permute(0, 2, 3, 1).contiguous().to(x.dtype)
            dw = dwn.permute(0, 2, 3, 1).contiguous().to(w.dtype)
        return dx, dw, None, None


def conv2d(x, w, stride: int = 1, pad: int = 0):
    return Conv2dFn.apply(x, w, stride, pad)


class LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps: float):
        x = x.contiguous()
        if _on_gpu(x):
            ops = require_hip()
            y, mean, rstd = ops.ln_fwd(x, weight, bias, eps)
        else:
            C = x.shape[-1]
            xf = x.float().reshape(-1, C)
            mean = xf.mean(dim=1)
            var = xf.var(dim=1, unbiased=False)
            rstd = (var + eps).rsqrt()
            y = ((xf - mean[:, None]) * rstd[:, None] * weight.float() +
                 bias.float()).to(x.dtype).reshape(x.shape)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        if _on_gpu(dy):
            ops = require_hip()
            dx, dw, db = ops.ln_bwd(x, dy, weight, mean, rstd)
        else:
            C = x.shape[-1]
            xf = x.float().reshape(-1, C)
            dyf = dy.float().reshape(-1, C)
            xhat = (xf - mean[:, None]) * rstd[:, None]
            dyg = dyf * weight.float()
            m1 = dyg.mean(dim=1, keepdim=True)
            m2 = (dyg * xhat).mean(dim=1, keepdim=True)
            dx = (rstd[:, None] * (dyg - m1 - xhat * m2)).to(x.dtype).reshape(x.shape)
            dw = (dyf * xhat).sum(0)
            db = dyf.sum(0)
        return dx, dw.to(weight.dtype), db.to(weight.dtype), None


def layer_norm(x, weight, bias, eps: float = 1e-5):
    return LayerNormFn.apply(x, weight, bias, eps)


class LayerNormAddFn(torch.autograd.Function):
    """y = LN(x + res) with the residual add fused into the norm's first
    pass (one kernel, z written once — removes the separate at::add the r1
    profiles showed on every BERT block). Backward fuses the symmetric
    grad: dx = dres = ln_bwd_dx; both inputs receive the same tensor."""

    @staticmethod
    def forward(ctx, x, res, weight, bias, eps: float):
        x = x.contiguous()
        res = res.contiguous()
        if _on_gpu(x):
            ops = require_hip()
            y, z, mean, rstd = ops.ln_add_fwd(x, res, weight, bias, eps)
        else:
            z = (x.float() + res.float()).to(x.dtype)
            C = x.shape[-1]
            zf = z.float().reshape(-1, C)
            mean = zf.mean(dim=1)
            var = zf.var(dim=1, unbiased=False)
            rstd = (var + eps).rsqrt()
            y = ((zf - mean[:, None]) * rstd[:, None] * weight.float() +
                 bias.float()).to(x.dtype).reshape(x.shape)
        ctx.save_for_backward(z, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        z, weight, mean, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        if _on_gpu(dy):
            dz, dw, db = require_hip().ln_bwd(z, dy, weight, mean, rstd)
        else:
            C = z.shape[-1]
            zf = z.float().reshape(-1, C)
            dyf = dy.float().reshape(-1, C)
            xhat = (zf - mean[:, None]) * rstd[:, None]
            dyg = dyf * weight.float()
            m1 = dyg.mean(dim=1, keepdim=True)
            m2 = (dyg * xhat).mean(dim=1, keepdim=True)
            dz = (rstd[:, None] * (dyg - m1 - xhat * m2)).to(z.dtype).reshape(z.shape)
            dw = (dyf * xhat).sum(0)
            db = dyf.sum(0)
        return dz, dz, dw.to(weight.dtype), db.to(weight.dtype), None


def layer_norm_add(x, res, weight, bias, eps: float = 1e-5):
    """LN(x + res) — fused residual-add + LayerNorm."""
    return LayerNormAddFn.apply(x, res, weight, bias, eps)


class BatchNormFn(torch.autograd.Function):
    """Training-mode BatchNorm over [*, C] (NHWC flattened), optional fused
    ReLU. Params/stats fp32; activations fp32 or bf16."""

    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var,
                momentum: float, eps: float, relu: bool):
        x = x.contiguous()
        if _on_gpu(x):
            ops = require_hip()
            rm = running_mean if running_mean is not None else torch.Tensor()
            rv = running_var if running_var is not None else torch.Tensor()
            y, mean, rstd = ops.bn_fwd_train(x, gamma, beta, rm, rv, momentum,
                                             eps, relu)
        else:
            C = x.shape[-1]
            xf = x.float().reshape(-1, C)
            mean = xf.mean(dim=0)
            var = xf.var(dim=0, unbiased=False)
            rstd = (var + eps).rsqrt()
            if running_mean is not None:
                M = xf.shape[0]
                unbiased = var * M / max(M - 1, 1)
                running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                running_var.mul_(1 - momentum).add_(unbiased, alpha=momentum)
            y = ((xf - mean) * rstd * gamma + beta)
            if relu:
                y = y.clamp_min(0)
            y = y.to(x.dtype).reshape(x.shape)
        ctx.save_for_backward(x, gamma, mean, rstd, y if relu else x)
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, rstd, y_post = ctx.saved_tensors
        dy = dy.contiguous()
        if _on_gpu(dy):
            ops = require_hip()
            dx, dgamma, dbeta = ops.bn_bwd(x, dy, y_post, mean, rstd, gamma,
                                           ctx.relu)
        else:
            C = x.shape[-1]
            xf = x.float().reshape(-1, C)
            dyf = dy.float().reshape(-1, C)
            if ctx.relu:
                mask = (y_post.float().reshape(-1, C) > 0).float()
                dyf = dyf * mask
            M = xf.shape[0]
            xhat = (xf - mean) * rstd
            sum_dy = dyf.sum(0)
            sum_dyx = (dyf * xhat).sum(0)
            dx = (rstd * gamma * (dyf - sum_dy / M - xhat * sum_dyx / M))
            dx = dx.to(x.dtype).reshape(x.shape)
            dgamma, dbeta = sum_dyx, sum_dy
        return dx, dgamma, dbeta, None, None, None, None, None


class BatchNormAddReLUFn(torch.autograd.Function):
    """y = relu(bn(x) + res) — the ResNet residual join fused into the BN
    normalize pass (saves the standalone add_relu kernel and one full
    activation read per block; backward writes dres alongside dx)."""

    @staticmethod
    def forward(ctx, x, res, gamma, beta, running_mean, running_var,
                momentum: float, eps: float):
        x = x.contiguous()
        res = res.contiguous()
        if _on_gpu(x):
            ops = require_hip()
            rm = running_mean if running_mean is not None else torch.Tensor()
            rv = running_var if running_var is not None else torch.Tensor()
            y, mean, rstd = ops.bn_fwd_train(x, gamma, beta, rm, rv, momentum,
                                             eps, True, res)
        else:
            C = x.shape[-1]
            xf = x.float().reshape(-1, C)
            mean = xf.mean(dim=0)
            var = xf.var(dim=0, unbiased=False)
            rstd = (var + eps).rsqrt()
            if running_mean is not None:
                M = xf.shape[0]
                unbiased = var * M / max(M - 1, 1)
                running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                running_var.mul_(1 - momentum).add_(unbiased, alpha=momentum)
            y = ((xf - mean) * rstd * gamma + beta +
                 res.float().reshape(-1, C)).clamp_min(0)
            y = y.to(x.dtype).reshape(x.shape)
        ctx.save_for_backward(x, gamma, mean, rstd, y)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, rstd, y_post = ctx.saved_tensors
        dy = dy.contiguous()
        if _on_gpu(dy):
            ops = require_hip()
            dx, dgamma, dbeta, dres = ops.bn_bwd(x, dy, y_post, mean, rstd,
                                                 gamma, True, True)
        else:
            C = x.shape[-1]
            xf = x.float().reshape(-1, C)
            dyf = dy.float().reshape(-1, C)
            mask = (y_post.float().reshape(-1, C) > 0).float()
            dyf = dyf * mask
            M = xf.shape[0]
            xhat = (xf - mean) * rstd
            sum_dy = dyf.sum(0)
            sum_dyx = (dyf * xhat).sum(0)
            dx = (rstd * gamma * (dyf - sum_dy / M - xhat * sum_dyx / M))
            dx = dx.to(x.dtype).reshape(x.shape)
            dres = dyf.to(x.dtype).reshape(x.shape)
            dgamma, dbeta = sum_dyx, sum_dy
        return dx, dres, dgamma, dbeta, None, None, None, None


def batch_norm_eval(x, gamma, beta, running_mean, running_var, eps: float,
                    relu: bool):
    x = x.contiguous()
    if _on_gpu(x):
        ops = require_hip()
        rstd = (running_var + eps).rsqrt()  # tiny glue op
        return ops.bn_fwd_eval(x, gamma, beta, running_mean, rstd, relu)
    C = x.shape[-1]
    xf = x.float().reshape(-1, C)
    y = (xf - running_mean) * (running_var + eps).rsqrt() * gamma + beta
    if relu:
        y = y.clamp_min(0)
    return y.to(x.dtype).reshape(x.shape)


class ReLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = x.contiguous()
        if _on_gpu(x):
            y = require_hip().relu_fwd(x)
        else:
            y = x.clamp_min(0)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        dy = dy.contiguous()
        if _on_gpu(dy):
            return require_hip().relu_bwd(dy, y)
        return dy * (y > 0).to(dy.dtype)


def relu(x):
    return ReLUFn.apply(x)


class AddReLUFn(torch.autograd.Function):
    """y = relu(a + b) — fused residual join (ResNet)."""

    @staticmethod
    def forward(ctx, a, b):
        a, b = a.contiguous(), b.contiguous()
        if _on_gpu(a):
            y = require_hip().add_relu_fwd(a, b)
        else:
            y = (a + b).clamp_min(0)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        dy = dy.contiguous()
        if _on_gpu(dy):
            da = require_hip().relu_bwd(dy, y)
        else:
            da = dy * (y > 0).to(dy.dtype)
        return da, da


def add_relu(a, b):
    return AddReLUFn.apply(a, b)


class AddScaledFn(torch.autograd.Function):
    """z = a + alpha * b — the LoRA combine (base + scaling * delta) as ONE
    kernel instead of a scale pass plus an add pass (HBM-bound glue)."""

    @staticmethod
    def forward(ctx, a, b, alpha):
        a, b = a.contiguous(), b.contiguous()
        ctx.alpha = alpha
        if _on_gpu(a):
            return require_hip().add_scaled_fwd(a, b, alpha)
        return a + alpha * b

    @staticmethod
    def backward(ctx, dz):
        dz = dz.contiguous()
        if _on_gpu(dz):
            db = require_hip().scale_fwd(dz, ctx.alpha)
        else:
            db = ctx.alpha * dz
        return dz, db, None


def add_scaled(a, b, alpha: float):
    return AddScaledFn.apply(a, b, alpha)


class LoraLinearFn(torch.autograd.Function):
    """The whole frozen-base LoRA linear as ONE node:
    y = x @ W^T + alpha * ((x @ A^T) @ B^T), with W frozen.

    Forward: the combine is the rank-r B GEMM's beta=1 C-accumulate
    epilogue writing into the base GEMM's output (no mark_dirty needed —
    the mutated tensor never escapes this node). Backward: dx is the base
    NT GEMM with the LoRA contribution beta=1-accumulated into it in the
    second GEMM's epilogue — at module level autograd would join the two
    branches with a separate elementwise add per adapter (the llama
    profile's dominant CUDAFunctor_add traffic); dA / dB^T are direct-NN
    skinny wgrads (only [M, r] operands transposed) with the tiny results
    scaled, never the big [M, out] grad."""

    @staticmethod
    def forward(ctx, x2, weight, lora_a, lora_b, alpha, weight_t):
        ctx.alpha = alpha
        if _on_gpu(x2):
            if _lib_gemm(x2):
                base = x2 @ weight.t()
                xa = x2 @ lora_a.t()
                # combine in the library GEMM's beta epilogue, in place
                y = base.addmm_(xa, lora_b.t(), alpha=alpha)
            else:
                ops = require_hip()
                base = ops.gemm(x2, weight, 0)
                xa = ops.gemm(x2, lora_a, 0)
                y = ops.gemm(xa, lora_b, 0, None, False, False, alpha, 1.0,
                             base)
        else:
            xa = x2 @ lora_a.t()
            y = x2 @ weight.t() + alpha * (xa @ lora_b.t())
        ctx.save_for_backward(x2, weight, lora_a, lora_b, xa, weight_t)
        return y

    @staticmethod
    def backward(ctx, dz):
        x2, weight, lora_a, lora_b, xa, weight_t = ctx.saved_tensors
        dz = dz.contiguous()
        alpha = ctx.alpha
        need_dx, _, need_da, need_db = ctx.needs_input_grad[:4]
        dx = da = db = None
        if _on_gpu(dz):
            if _lib_gemm(dz):
                d_xa = (dz @ lora_b).mul_(alpha)       # [M, r] tiny scale
                if need_da:
                    da = (d_xa.t() @ x2).to(lora_a.dtype)
                if need_db:
                    db = (dz.t() @ xa).mul_(alpha).to(lora_b.dtype)
                if need_dx:
                    dx = dz @ weight
                    dx = dx.addmm_(d_xa, lora_a)       # lora dx in epilogue
                return dx, None, da, db, None, None
            ops = require_hip()
            d_xa = ops.scale_fwd(ops.gemm(dz, lora_b, 1), alpha)  # [M, r]
            if need_da:
                da = ops.gemm(d_xa.t().contiguous(), x2, 1, None, False,
                              False, 1.0, 0.0, None, True).to(lora_a.dtype)
            if need_db:
                dbt = ops.gemm(xa.t().contiguous(), dz, 1, None, False,
                               False, 1.0, 0.0, None, True)    # [r, out]
                db = ops.scale_fwd(dbt, alpha).t().contiguous()
            if need_dx:
                if weight_t is not None:
                    dx = ops.gemm(dz, weight_t, 0)     # NT on cached W^T
                else:
                    dx = ops.gemm(dz, weight, 1)       # NN: dz @ W
                # one explicit in-place add (vs autograd's AccumulateGrad
                # pass); a beta=1 epilogue join on this K'=16 NN measured
                # SLOWER than the add it replaces (38.0 vs 38.7 samples/s)
                dx = dx.add_(ops.gemm(d_xa, lora_a, 1))
        else:
            d_xa = alpha * (dz.float() @ lora_b.float())
            if need_da:
                da = (d_xa.t() @ x2.float()).to(lora_a.dtype)
            if need_db:
                db = alpha * (dz.float().t() @ xa.float()).to(lora_b.dtype)
            if need_dx:
                dx = (dz.float() @ weight.float()
                      + d_xa @ lora_a.float()).to(dz.dtype)
        return dx, None, da, db, None, None


def lora_linear(x, weight, lora_a, lora_b, scaling: float, weight_t=None):
    """Frozen-base LoRA linear: base + rank-r pair + combine as one
    autograd node (LoraLinearFn) — fused epilogues fwd AND bwd."""
    shape = x.shape
    x2 = x.reshape(-1, shape[-1]).contiguous()
    y = LoraLinearFn.apply(x2, weight, lora_a, lora_b, scaling, weight_t)
    return y.reshape(*shape[:-1], weight.shape[0])


class GELUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = x.contiguous()
        ctx.save_for_backward(x)
        if _on_gpu(x):
            return require_hip().gelu_fwd(x)
        return F.gelu(x.float(), approximate="tanh").to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        dy = dy.contiguous()
        if _on_gpu(dy):
            return require_hip().gelu_bwd(dy, x)
        xf = x.float().detach().requires_grad_(True)
        with torch.enable_grad():
            y = F.gelu(xf, approximate="tanh")
        (g,) = torch.autograd.grad(y, xf, dy.float())
        return g.to(x.dtype)


def gelu(x):
    return GELUFn.apply(x)


class MSELossFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, target):
        x, target = x.contiguous(), target.contiguous()
        ctx.save_for_backward(x, target)
        if _on_gpu(x):
            return require_hip().mse_fwd(x, target)
        return F.mse_loss(x.float(), target.float())

    @staticmethod
    def backward(ctx, dout):
        x, target = ctx.saved_tensors
        if _on_gpu(x):
            dx = require_hip().mse_bwd(x, target, dout.float().contiguous())
        else:
            dx = 2.0 * (x.float() - target.float()) / x.numel() * dout.float()
            dx = dx.to(x.dtype)
        return dx, None


def mse_loss(x, target):
    return MSELossFn.apply(x, target)


class CrossEntropyFn(torch.autograd.Function):
    """Fused log-softmax + NLL over [B, C] logits, int64 targets, mean
    reduction."""

    @staticmethod
    def forward(ctx, logits, target):
        logits, target = logits.contiguous(), target.contiguous()
        if _on_gpu(logits):
            loss, lse = require_hip().ce_fwd(logits, target)
        else:
            lse = torch.logsumexp(logits.float(), dim=1)
            loss = (lse - logits.float().gather(
                1, target[:, None]).squeeze(1)).mean()
        ctx.save_for_backward(logits, target, lse)
        return loss

    @staticmethod
    def backward(ctx, dout):
        logits, target, lse = ctx.saved_tensors
        if _on_gpu(logits):
            dx = require_hip().ce_bwd(logits, target, lse,
                                      dout.float().contiguous())
        else:
            p = (logits.float() - lse[:, None]).exp()
            p.scatter_add_(1, target[:, None],
                           torch.full_like(target[:, None], -1.0,
                                           dtype=torch.float32))
            dx = (p * dout.float() / logits.shape[0]).to(logits.dtype)
        return dx, None


def cross_entropy(logits, target):
    return CrossEntropyFn.apply(logits, target)


class BatchedMatmulFn(torch.autograd.Function):
    """Batched GEMM on MFMA (grid.z = batch) for the attention path.
    layout 0 (NT): C = A @ B^T   A:[nb,M,K] B:[nb/g,N,K]
    layout 1 (NN): C = A @ B     A:[nb,M,K] B:[nb/g,K,N]
    b_group = g > 1 is GQA: g consecutive A batches (query heads) share one
    B batch (KV head) — the kernel indexes B by z/g, so no repeated KV copy
    is ever materialized. alpha scales the product."""

    @staticmethod
    def forward(ctx, A, B, layout: int, alpha: float = 1.0, b_group: int = 1):
        A, B = A.contiguous(), B.contiguous()
        ctx.save_for_backward(A, B)
        ctx.layout, ctx.alpha, ctx.b_group = layout, alpha, b_group
        if _on_gpu(A):
            return require_hip().gemm_batched(A, B, layout, False, alpha, b_group)
        Bx = B.float().repeat_interleave(b_group, dim=0) if b_group > 1 else B.float()
        if layout == 0:
            return torch.bmm(A.float(), Bx.transpose(1, 2)).mul(alpha).to(A.dtype)
        return torch.bmm(A.float(), Bx).mul(alpha).to(A.dtype)

    @staticmethod
    def backward(ctx, dC):
        A, B = ctx.saved_tensors
        layout, alpha, g = ctx.layout, ctx.alpha, ctx.b_group
        dC = dC.contiguous()
        if _on_gpu(dC):
            ops = require_hip()
            if layout == 0:   # C = A@B^T : dA = dC@B (NN), dB = dC^T@A (TN)
                dA = ops.gemm_batched(dC, B, 1, False, alpha, g)
                dBf = ops.gemm_batched(dC, A, 2, False, alpha)
            else:             # C = A@B  : dA = dC@B^T (NT), dB = A^T@dC (TN)
                dA = ops.gemm_batched(dC, B, 0, False, alpha, g)
                dBf = ops.gemm_batched(A, dC, 2, False, alpha)
        else:
            Bx = B.float().repeat_interleave(g, dim=0) if g > 1 else B.float()
            if layout == 0:
                dA = torch.bmm(dC.float(), Bx).mul(alpha).to(A.dtype)
                dBf = torch.bmm(dC.float().transpose(1, 2), A.float()).mul(alpha).to(B.dtype)
            else:
                dA = torch.bmm(dC.float(), Bx.transpose(1, 2)).mul(alpha).to(A.dtype)
                dBf = torch.bmm(A.float().transpose(1, 2), dC.float()).mul(alpha).to(B.dtype)
        if g > 1:  # sum query-head contributions back onto the shared KV head
            nb = dBf.shape[0] // g
            dB = dBf.reshape(nb, g, *dBf.shape[1:]).float().sum(1).to(B.dtype)
        else:
            dB = dBf
        return dA, dB.to(B.dtype), None, None, None


def batched_matmul(A, B, layout: int, alpha: float = 1.0, b_group: int = 1):
    return BatchedMatmulFn.apply(A, B, layout, alpha, b_group)


class SoftmaxFn(torch.autograd.Function):
    """y = softmax(scale * x [+ causal mask]) over the last dim."""

    @staticmethod
    def forward(ctx, x, scale: float, causal_seq: int = 0):
        x = x.contiguous()
        if _on_gpu(x):
            y = require_hip().softmax_fwd(x, scale, causal_seq)
        else:
            z = x.float() * scale
            if causal_seq > 0:
                S = causal_seq
                C = x.shape[-1]
                q = torch.arange(z.numel() // C) % S
                mask = torch.arange(C)[None, :] > q[:, None]
                z = z.reshape(-1, C).masked_fill(mask, float("-inf")).reshape(x.shape)
            y = torch.softmax(z, dim=-1).to(x.dtype)
        ctx.save_for_backward(y)
        ctx.scale = scale
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        dy = dy.contiguous()
        if _on_gpu(dy):
            return require_hip().softmax_bwd(y, dy, ctx.scale), None, None
        yf = y.float()
        dyf = dy.float()
        dot = (yf * dyf).sum(-1, keepdim=True)
        return (ctx.scale * yf * (dyf - dot)).to(y.dtype), None, None


def softmax(x, scale: float = 1.0, causal_seq: int = 0):
    return SoftmaxFn.apply(x, scale, causal_seq)


def attention(q, k, v, causal: bool = False):
    """Multi-head attention core on batched MFMA GEMMs + fused softmax.
    q: [nb, S, Dh] (nb = B*H); k, v: [nb, S, Dh] or [nb/g, S, Dh] (GQA —
    the shared KV heads are indexed in-kernel, never copied)."""
    import math

    S, Dh = q.shape[-2], q.shape[-1]
    g = q.shape[0] // k.shape[0]
    scores = batched_matmul(q, k, 0, 1.0, g)              # [nb, S, S]
    probs = softmax(scores, 1.0 / math.sqrt(Dh), S if causal else 0)
    return batched_matmul(probs, v, 1, 1.0, g)            # [nb, S, Dh]


def _attn_core_fwd(ops, q, k, v, causal):
    """q/k/v: [B,S,h|kvh,dh] strided VIEWS — consumed in place by the
    strided batched GEMM (no permute copies). Returns (o [B,S,h,dh]
    contiguous-in-that-layout, probs [B*h,S,S])."""
    import math

    B, S, h, dh = q.shape
    kvh = k.shape[2]
    g = h // kvh
    nb = B * h
    scores = ops.bmm_strided(
        q, k, None, layout=0, M=S, N=S, K=dh, nbatch=nb, heads=h, b_group=g,
        alpha=1.0, saO=q.stride(0), saI=q.stride(2), lda=q.stride(1),
        sbO=k.stride(0), sbI=k.stride(2), ldb=k.stride(1))
    probs = ops.softmax_fwd(scores, 1.0 / math.sqrt(dh), S if causal else 0)
    o = torch.empty(B, S, h, dh, dtype=q.dtype, device=q.device)
    ops.bmm_strided(
        probs, v, o, layout=1, M=S, N=dh, K=S, nbatch=nb, heads=h, b_group=g,
        alpha=1.0, saO=h * S * S, saI=S * S, lda=S,
        sbO=v.stride(0), sbI=v.stride(2), ldb=v.stride(1),
        scO=o.stride(0), scI=o.stride(2), ldc=o.stride(1))
    return o, probs


def _attn_core_bwd(ops, q, k, v, probs, do, dq, dk_qh, dv_qh):
    """Backward of _attn_core_fwd. do: [B,S,h,dh]; dq/dk_qh/dv_qh are
    PER-Q-HEAD [B,S,h,dh] destinations written in place via strided C
    (GQA callers group-sum dk_qh/dv_qh afterwards)."""
    import math

    B, S, h, dh = do.shape
    kvh = k.shape[2]
    g = h // kvh
    nb = B * h
    dP = ops.bmm_strided(
        do, v, None, layout=0, M=S, N=S, K=dh, nbatch=nb, heads=h, b_group=g,
        alpha=1.0, saO=do.stride(0), saI=do.stride(2), lda=do.stride(1),
        sbO=v.stride(0), sbI=v.stride(2), ldb=v.stride(1))
    dS = ops.softmax_bwd(probs, dP, 1.0 / math.sqrt(dh))
    ops.bmm_strided(   # dq = dS @ k (NN, GQA-shared k)
        dS, k, dq, layout=1, M=S, N=dh, K=S, nbatch=nb, heads=h, b_group=g,
        alpha=1.0, saO=h * S * S, saI=S * S, lda=S,
        sbO=k.stride(0), sbI=k.stride(2), ldb=k.stride(1),
        scO=dq.stride(0), scI=dq.stride(2), ldc=dq.stride(1))
    ops.bmm_strided(   # dk = dS^T @ q (TN, per q-head)
        dS, q, dk_qh, layout=2, M=S, N=dh, K=S, nbatch=nb, heads=h, b_group=1,
        alpha=1.0, saO=h * S * S, saI=S * S, lda=S,
        sbO=q.stride(0), sbI=q.stride(2), ldb=q.stride(1),
        scO=dk_qh.stride(0), scI=dk_qh.stride(2), ldc=dk_qh.stride(1))
    ops.bmm_strided(   # dv = P^T @ dO (TN, per q-head)
        probs, do, dv_qh, layout=2, M=S, N=dh, K=S, nbatch=nb, heads=h,
        b_group=1, alpha=1.0, saO=h * S * S, saI=S * S, lda=S,
        sbO=do.stride(0), sbI=do.stride(2), ldb=do.stride(1),
        scO=dv_qh.stride(0), scI=dv_qh.stride(2), ldc=dv_qh.stride(1))


def _flash_eligible(q, k, v) -> bool:
    """Fused flash path: bf16, dh in {32,64,128}, contiguous head_dim.
    BATON_NO_FLASH=1 forces the materialized-scores path (A/B runs)."""
    import os

    return (
        q.dtype == torch.bfloat16
        and q.shape[-1] in (32, 64, 128)
        and q.stride(-1) == 1 and k.stride(-1) == 1 and v.stride(-1) == 1
        and os.environ.get("BATON_NO_FLASH", "0") != "1"
    )


class AttnPackedFn(torch.autograd.Function):
    """Attention over a PACKED qkv [B,S,3,h,dh] (BERT-style fused QKV):
    q/k/v are consumed as views, backward writes straight into one dqkv —
    zero permute/assembly copies on either pass. Fused flash kernels
    (flash.hip) when eligible; materialized-scores path otherwise."""

    @staticmethod
    def forward(ctx, qkv, causal):
        ops = require_hip()
        q, k, v = qkv[:, :, 0], qkv[:, :, 1], qkv[:, :, 2]
        ctx.causal = causal
        if _flash_eligible(q, k, v):
            B, S, h, dh = q.shape
            o = torch.empty(B, S, h, dh, dtype=q.dtype, device=q.device)
            _, lse = ops.flash_fwd(q, k, v, o, causal)
            ctx.save_for_backward(qkv, o, lse)
            ctx.flash = True
            return o
        o, probs = _attn_core_fwd(ops, q, k, v, causal)
        ctx.save_for_backward(qkv, probs)
        ctx.flash = False
        return o

    @staticmethod
    def backward(ctx, do):
        ops = require_hip()
        do = do.contiguous()
        if ctx.flash:
            qkv, o, lse = ctx.saved_tensors
            q, k, v = qkv[:, :, 0], qkv[:, :, 1], qkv[:, :, 2]
            dqkv = torch.empty_like(qkv)
            ops.flash_bwd(q, k, v, o, do, lse,
                          dqkv[:, :, 0], dqkv[:, :, 1], dqkv[:, :, 2],
                          ctx.causal)
            return dqkv, None
        qkv, probs = ctx.saved_tensors
        dqkv = torch.empty_like(qkv)
        q, k, v = qkv[:, :, 0], qkv[:, :, 1], qkv[:, :, 2]
        _attn_core_bwd(ops, q, k, v, probs, do,
                       dqkv[:, :, 0], dqkv[:, :, 1], dqkv[:, :, 2])
        return dqkv, None


def attention_qkv(qkv, causal: bool = False):
    """qkv: [B,S,3,h,dh] (one fused-projection tensor) -> o [B,S,h,dh].
    Packed layout implies equal q/k/v head counts (no GQA) — use
    :func:`attention_bshd` with separate k/v views for grouped KV."""
    assert qkv.dim() == 5 and qkv.shape[2] == 3, "qkv must be [B,S,3,h,dh]"
    if _on_gpu(qkv):
        return AttnPackedFn.apply(qkv, causal)
    B, S, _, h, dh = qkv.shape
    q, k, v = (qkv[:, :, i].permute(0, 2, 1, 3).reshape(B * h, S, dh)
               for i in range(3))
    o = attention(q, k, v, causal)
    return o.reshape(B, h, S, dh).permute(0, 2, 1, 3)


class AttnBSHDFn(torch.autograd.Function):
    """Attention over separate q [B,S,h,dh], k/v [B,S,kvh,dh] (Llama GQA:
    kv heads stay un-replicated AND un-permuted — strided views)."""

    @staticmethod
    def forward(ctx, q, k, v, causal):
        ops = require_hip()
        ctx.causal = causal
        if _flash_eligible(q, k, v):
            B, S, h, dh = q.shape
            o = torch.empty(B, S, h, dh, dtype=q.dtype, device=q.device)
            _, lse = ops.flash_fwd(q, k, v, o, causal)
            ctx.save_for_backward(q, k, v, o, lse)
            ctx.flash = True
            return o
        o, probs = _attn_core_fwd(ops, q, k, v, causal)
        ctx.save_for_backward(q, k, v, probs)
        ctx.flash = False
        return o

    @staticmethod
    def backward(ctx, do):
        ops = require_hip()
        do = do.contiguous()
        B, S, h, dh = do.shape
        dq = torch.empty(B, S, h, dh, dtype=do.dtype, device=do.device)
        dk_qh = torch.empty_like(dq)
        dv_qh = torch.empty_like(dq)
        if ctx.flash:
            q, k, v, o, lse = ctx.saved_tensors
            ops.flash_bwd(q, k, v, o, do, lse, dq, dk_qh, dv_qh, ctx.causal)
        else:
            q, k, v, probs = ctx.saved_tensors
            _attn_core_bwd(ops, q, k, v, probs, do, dq, dk_qh, dv_qh)
        kvh = k.shape[2]
        if kvh != h:
            g = h // kvh
            dk = dk_qh.view(B, S, kvh, g, dh).sum(3, dtype=torch.float32).to(do.dtype)
            dv = dv_qh.view(B, S, kvh, g, dh).sum(3, dtype=torch.float32).to(do.dtype)
        else:
            dk, dv = dk_qh, dv_qh
        return dq, dk, dv, None


def attention_bshd(q, k, v, causal: bool = False):
    """q: [B,S,h,dh]; k/v: [B,S,kvh,dh] (views allowed) -> o [B,S,h,dh]."""
    if _on_gpu(q):
        return AttnBSHDFn.apply(q, k, v, causal)
    B, S, h, dh = q.shape
    kvh = k.shape[2]
    qf = q.permute(0, 2, 1, 3).reshape(B * h, S, dh)
    kf = k.permute(0, 2, 1, 3).reshape(B * kvh, S, dh)
    vf = v.permute(0, 2, 1, 3).reshape(B * kvh, S, dh)
    o = attention(qf, kf, vf, causal)
    return o.reshape(B, h, S, dh).permute(0, 2, 1, 3)


class RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps: float):
        x = x.contiguous()
        if _on_gpu(x):
            y, rstd = require_hip().rms_fwd(x, weight, eps)
        else:
            C = x.shape[-1]
            xf = x.float().reshape(-1, C)
            rstd = (xf.pow(2).mean(dim=1) + eps).rsqrt()
            y = (xf * rstd[:, None] * weight.float()).to(x.dtype).reshape(x.shape)
        ctx.save_for_backward(x, weight, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        if _on_gpu(dy):
            dx, dw = require_hip().rms_bwd(x, dy, weight, rstd)
        else:
            C = x.shape[-1]
            xf = x.float().reshape(-1, C)
            dyf = dy.float().reshape(-1, C)
            xhat = xf * rstd[:, None]
            dyw = dyf * weight.float()
            m = (dyw * xhat).mean(dim=1, keepdim=True)
            dx = (rstd[:, None] * (dyw - xhat * m)).to(x.dtype).reshape(x.shape)
            dw = (dyf * xhat).sum(0)
        return dx, dw.to(weight.dtype), None


def rms_norm(x, weight, eps: float = 1e-5):
    return RMSNormFn.apply(x, weight, eps)


class AddRMSNormFn(torch.autograd.Function):
    """(y, z) = (RMS(x + res) * w, x + res): the transformer residual add
    fused into the norm's square-sum pass. z is a REAL output (the llama
    residual stream threads through it), so the next block's add fuses
    too. Backward fuses the residual-stream gradient into dx via the
    PLUS kernel variant (rms_bwd_plus) — no separate add pass either way."""

    @staticmethod
    def forward(ctx, x, res, weight, eps: float):
        x = x.contiguous()
        res = res.contiguous()
        if _on_gpu(x):
            y, z, rstd = require_hip().rms_add_fwd(x, res, weight, eps)
        else:
            z = (x.float() + res.float()).to(x.dtype)
            C = x.shape[-1]
            zf = z.float().reshape(-1, C)
            rstd = (zf.pow(2).mean(dim=1) + eps).rsqrt()
            y = (zf * rstd[:, None] * weight.float()).to(x.dtype).reshape(x.shape)
        ctx.save_for_backward(z, weight, rstd)
        return y, z

    @staticmethod
    def backward(ctx, dy, dz):
        z, weight, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        if _on_gpu(dy):
            ops = require_hip()
            if dz is not None:
                dx, dw = ops.rms_bwd_plus(z, dy, weight, rstd, dz.contiguous())
            else:
                dx, dw = ops.rms_bwd(z, dy, weight, rstd)
        else:
            C = z.shape[-1]
            zf = z.float().reshape(-1, C)
            dyf = dy.float().reshape(-1, C)
            xhat = zf * rstd[:, None]
            dyw = dyf * weight.float()
            m = (dyw * xhat).mean(dim=1, keepdim=True)
            dx = (rstd[:, None] * (dyw - xhat * m)).reshape(z.shape)
            if dz is not None:
                dx = dx + dz.float()
            dx = dx.to(z.dtype)
            dw = (dyf * xhat).sum(0)
        return dx, dx, dw.to(weight.dtype), None


def add_rms_norm(x, res, weight, eps: float = 1e-5):
    """(normed, z) = (RMS(x + res) * w, x + res)."""
    return AddRMSNormFn.apply(x, res, weight, eps)


def rope_tables(seq_len: int, head_dim: int, base: float = 500000.0,
                device=None) -> tuple:
    """Host-precomputed RoPE cos/sin tables [S, D/2] fp32 (Llama-3 base
    5e5). Trig stays off the GPU hot path (guide Appendix B)."""
    half = head_dim // 2
    inv_freq = 1.0 / (base ** (torch.arange(half, dtype=torch.float32) / half))
    t = torch.arange(seq_len, dtype=torch.float32)
    ang = torch.outer(t, inv_freq)
    cos, sin = ang.cos(), ang.sin()
    if device is not None:
        cos, sin = cos.to(device), sin.to(device)
    return cos, sin


class RoPEFn(torch.autograd.Function):
    """Neox-style half-rotation on [B, S, H, D]."""

    @staticmethod
    def forward(ctx, x, cos_t, sin_t):
        x = x.contiguous()
        ctx.save_for_backward(cos_t, sin_t)
        if _on_gpu(x):
            return require_hip().rope(x, cos_t, sin_t, False)
        return RoPEFn._cpu(x, cos_t, sin_t, inverse=False)

    @staticmethod
    def backward(ctx, dy):
        cos_t, sin_t = ctx.saved_tensors
        dy = dy.contiguous()
        if _on_gpu(dy):
            return require_hip().rope(dy, cos_t, sin_t, True), None, None
        return RoPEFn._cpu(dy, cos_t, sin_t, inverse=True), None, None

    @staticmethod
    def _cpu(x, cos_t, sin_t, inverse):
        B, S, H, D = x.shape
        half = D // 2
        xf = x.float()
        x1, x2 = xf[..., :half], xf[..., half:]
        c = cos_t[:S, None, :].to(x.device)
        s = sin_t[:S, None, :].to(x.device)
        if inverse:
            o1 = x1 * c + x2 * s
            o2 = -x1 * s + x2 * c
        else:
            o1 = x1 * c - x2 * s
            o2 = x1 * s + x2 * c
        return torch.cat([o1, o2], dim=-1).to(x.dtype)


def rope(x, cos_t, sin_t):
    return RoPEFn.apply(x, cos_t, sin_t)


class SiluMulFn(torch.autograd.Function):
    """SwiGLU gate: y = silu(a) * b."""

    @staticmethod
    def forward(ctx, a, b):
        a, b = a.contiguous(), b.contiguous()
        ctx.save_for_backward(a, b)
        if _on_gpu(a):
            return require_hip().silu_mul_fwd(a, b)
        af = a.float()
        return (af * torch.sigmoid(af) * b.float()).to(a.dtype)

    @staticmethod
    def backward(ctx, dy):
        a, b = ctx.saved_tensors
        dy = dy.contiguous()
        if _on_gpu(dy):
            da, db = require_hip().silu_mul_bwd(dy, a, b)
            return da, db
        af, bf, dyf = a.float(), b.float(), dy.float()
        sig = torch.sigmoid(af)
        silu = af * sig
        dsilu = sig * (1 + af * (1 - sig))
        return (dyf * bf * dsilu).to(a.dtype), (dyf * silu).to(b.dtype)


def silu_mul(a, b):
    return SiluMulFn.apply(a, b)
