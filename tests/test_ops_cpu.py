"""CPU-path tests of the op layer: the autograd.Function wiring (fallback
branch) against stock torch modules, so the wrappers are verified even
without a GPU. The HIP branch of the same Functions is covered by
tests/test_kernels_gpu.py on the GPU box."""

import pytest
import torch

from baton_amd.ops import functional as BF
from baton_amd.ops.modules import (
    BatonBatchNorm2d,
    BatonConv2d,
    BatonLayerNorm,
    BatonLinear,
)


def test_linear_matches_torch():
    torch.manual_seed(0)
    lin = BatonLinear(16, 8)
    ref = torch.nn.Linear(16, 8)
    with torch.no_grad():
        ref.weight.copy_(lin.weight)
        ref.bias.copy_(lin.bias)
    x = torch.randn(4, 16, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    y = lin(x)
    yr = ref(x2)
    assert torch.allclose(y, yr, atol=1e-6)
    y.sum().backward()
    yr.sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)
    assert torch.allclose(lin.weight.grad, ref.weight.grad, atol=1e-6)
    assert torch.allclose(lin.bias.grad.float(), ref.bias.grad, atol=1e-6)


def test_conv2d_matches_torch():
    torch.manual_seed(1)
    conv = BatonConv2d(8, 16, 3, stride=2, padding=1)
    x = torch.randn(2, 10, 10, 8, requires_grad=True)
    y = conv(x)
    # reference
    xr = x.detach().permute(0, 3, 1, 2).clone().requires_grad_(True)
    wr = conv.weight.detach().permute(0, 3, 1, 2).clone().requires_grad_(True)
    yr = torch.nn.functional.conv2d(xr, wr, stride=2, padding=1)
    assert torch.allclose(y, yr.permute(0, 2, 3, 1), atol=1e-5)
    dy = torch.randn_like(y)
    y.backward(dy)
    yr.backward(dy.permute(0, 3, 1, 2))
    assert torch.allclose(x.grad, xr.grad.permute(0, 2, 3, 1), atol=1e-5)
    assert torch.allclose(conv.weight.grad, wr.grad.permute(0, 2, 3, 1), atol=1e-4)


def test_layernorm_matches_torch():
    torch.manual_seed(2)
    ln = BatonLayerNorm(32)
    ref = torch.nn.LayerNorm(32)
    with torch.no_grad():
        ln.weight.copy_(ref.weight)
        ln.bias.copy_(ref.bias)
    x = torch.randn(6, 32, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    y = ln(x)
    yr = ref(x2)
    assert torch.allclose(y, yr, atol=1e-5)
    dy = torch.randn_like(y)
    y.backward(dy)
    yr.backward(dy)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(ln.weight.grad, ref.weight.grad, atol=1e-4)


def test_batchnorm_matches_torch():
    torch.manual_seed(3)
    bn = BatonBatchNorm2d(8)
    ref = torch.nn.BatchNorm2d(8)
    x = torch.randn(4, 5, 5, 8, requires_grad=True)
    x2 = x.detach().permute(0, 3, 1, 2).clone().requires_grad_(True)
    bn.train(); ref.train()
    y = bn(x)
    yr = ref(x2)
    assert torch.allclose(y, yr.permute(0, 2, 3, 1), atol=1e-5)
    assert torch.allclose(bn.running_mean, ref.running_mean, atol=1e-6)
    assert torch.allclose(bn.running_var, ref.running_var, atol=1e-5)
    dy = torch.randn_like(y)
    y.backward(dy)
    yr.backward(dy.permute(0, 3, 1, 2))
    assert torch.allclose(x.grad, x2.grad.permute(0, 2, 3, 1), atol=1e-5)
    assert torch.allclose(bn.weight.grad, ref.weight.grad, atol=1e-4)
    # eval path
    bn.eval(); ref.eval()
    ye = bn(x.detach())
    yre = ref(x.detach().permute(0, 3, 1, 2))
    assert torch.allclose(ye, yre.permute(0, 2, 3, 1), atol=1e-5)


def test_fused_bn_relu():
    torch.manual_seed(4)
    bn = BatonBatchNorm2d(4, fused_relu=True)
    x = torch.randn(8, 3, 3, 4, requires_grad=True)
    y = bn(x)
    assert (y >= 0).all()
    y.sum().backward()
    assert x.grad is not None


def test_add_relu_and_gelu_grads():
    torch.manual_seed(5)
    a = torch.randn(50, requires_grad=True)
    b = torch.randn(50, requires_grad=True)
    y = BF.add_relu(a, b)
    assert torch.allclose(y, (a + b).clamp_min(0))
    y.sum().backward()
    mask = ((a + b) > 0).float()
    assert torch.allclose(a.grad, mask)
    assert torch.allclose(b.grad, mask)

    x = torch.randn(50, requires_grad=True)
    y = BF.gelu(x)
    dy = torch.randn(50)
    y.backward(dy)
    xr = x.detach().requires_grad_(True)
    torch.nn.functional.gelu(xr, approximate="tanh").backward(dy)
    assert torch.allclose(x.grad, xr.grad, atol=1e-5)


def test_add_scaled_grads():
    torch.manual_seed(15)
    a = torch.randn(64, requires_grad=True)
    b = torch.randn(64, requires_grad=True)
    z = BF.add_scaled(a, b, 0.25)
    assert torch.allclose(z, a + 0.25 * b)
    dz = torch.randn(64)
    z.backward(dz)
    assert torch.allclose(a.grad, dz)
    assert torch.allclose(b.grad, 0.25 * dz)


def test_lora_join_grads():
    """Fused LoRA combine (in-place C-accumulate join) vs a plain autograd
    composition: values and all three grads, including the pass-through
    into the base GEMM's input."""
    torch.manual_seed(16)
    M, K, r, N = 12, 8, 4, 10
    x = torch.randn(M, K)
    w = torch.randn(N, K)           # frozen base
    a = torch.randn(r, K, requires_grad=True)
    b = torch.randn(N, r, requires_grad=True)
    scaling = 0.5

    y = BF.lora_linear(x, w, a, b, scaling)
    dz = torch.randn(M, N)
    y.backward(dz)

    ar = a.detach().requires_grad_(True)
    br = b.detach().requires_grad_(True)
    yr = x @ w.t() + scaling * ((x @ ar.t()) @ br.t())
    yr.backward(dz)
    assert torch.allclose(y, yr, atol=1e-5)
    assert torch.allclose(a.grad, ar.grad, atol=1e-5)
    assert torch.allclose(b.grad, br.grad, atol=1e-5)

    # 3-D input path (B, S, K) and x needing grad through both branches
    x3 = torch.randn(2, 6, K, requires_grad=True)
    x3r = x3.detach().requires_grad_(True)
    y3 = BF.lora_linear(x3, w, a, b, scaling)
    y3r = x3r @ w.t() + scaling * ((x3r @ a.t()) @ b.t())
    g3 = torch.randn_like(y3)
    y3.backward(g3)
    y3r.backward(g3)
    assert torch.allclose(y3, y3r, atol=1e-5)
    assert torch.allclose(x3.grad, x3r.grad, atol=1e-5)


def test_losses_match_torch():
    torch.manual_seed(6)
    x = torch.randn(20, 1, requires_grad=True)
    t = torch.randn(20, 1)
    loss = BF.mse_loss(x, t)
    assert torch.allclose(loss, torch.nn.functional.mse_loss(x, t), atol=1e-6)
    loss.backward()
    assert torch.allclose(x.grad, 2 * (x.detach() - t) / x.numel(), atol=1e-6)

    logits = torch.randn(16, 10, requires_grad=True)
    target = torch.randint(0, 10, (16,))
    l2 = BF.cross_entropy(logits, target)
    ref = torch.nn.functional.cross_entropy(logits, target)
    assert torch.allclose(l2, ref, atol=1e-5)
    l2.backward()
    lr = logits.detach().requires_grad_(True)
    torch.nn.functional.cross_entropy(lr, target).backward()
    assert torch.allclose(logits.grad, lr.grad, atol=1e-5)


def test_fused_optimizers_match_torch_cpu():
    from baton_amd.ops.optim import FusedAdam, FusedSGD

    torch.manual_seed(7)
    for make_opt, make_ref in [
        (lambda ps: FusedSGD(ps, lr=0.05, momentum=0.9, weight_decay=1e-4),
         lambda ps: torch.optim.SGD(ps, lr=0.05, momentum=0.9, weight_decay=1e-4)),
        (lambda ps: FusedAdam(ps, lr=0.01, weight_decay=1e-3),
         lambda ps: torch.optim.Adam(ps, lr=0.01, weight_decay=1e-3)),
    ]:
        m1 = torch.nn.Linear(10, 10)
        m2 = torch.nn.Linear(10, 10)
        m2.load_state_dict(m1.state_dict())
        o1, o2 = make_opt(m1.parameters()), make_ref(m2.parameters())
        for _ in range(5):
            x = torch.randn(8, 10)
            o1.zero_grad(); o2.zero_grad()
            m1(x).pow(2).mean().backward()
            m2(x).pow(2).mean().backward()
            o1.step(); o2.step()
        for p1, p2 in zip(m1.parameters(), m2.parameters()):
            assert torch.allclose(p1, p2, atol=1e-5), "optimizer trajectory diverged"


def test_resnet18_cpu_smoke():
    from baton_amd.models.resnet import make_synthetic_cifar, resnet18

    torch.manual_seed(8)
    m = resnet18()
    x, y = make_synthetic_cifar(8)
    hist = m.train_round(x, y, n_epoch=2)
    assert len(hist) == 2
    assert all(v == v for v in hist)
    assert hist[1] < hist[0] + 0.5
