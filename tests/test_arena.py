"""FlatParamArena invariants (runtime/arena.py) — dtype-grouped flats."""

import pytest
import torch
import torch.nn as nn

from baton_amd.runtime.arena import FlatParamArena


def make_model():
    torch.manual_seed(0)
    return nn.Sequential(
        nn.Linear(8, 16), nn.ReLU(), nn.BatchNorm1d(16), nn.Linear(16, 4)
    )


def test_params_alias_flat():
    m = make_model()
    before = {n: p.detach().clone() for n, p in m.named_parameters()}
    arena = FlatParamArena(m)
    assert len(arena.groups) == 1  # all fp32
    assert arena.check_views()
    for n, p in m.named_parameters():
        assert torch.equal(p.detach(), before[n])
    arena.groups[0].flat.zero_()
    for p in m.parameters():
        assert p.detach().abs().sum() == 0


def test_backward_accumulates_into_flat_grads():
    m = make_model()
    arena = FlatParamArena(m)
    m(torch.randn(4, 8)).sum().backward()
    g = arena.groups[0]
    assert g.grad.abs().sum() > 0
    for n, p in m.named_parameters():
        lo, hi = g.slices[n]
        assert p.grad.data_ptr() == g.grad[lo:hi].data_ptr()
    arena.zero_grads()
    assert g.grad.abs().sum() == 0


def test_mixed_dtype_groups():
    """bf16 weights + fp32 BN params produce two groups (the ResNet case)."""
    m = make_model()
    m[0] = m[0].to(torch.bfloat16)
    m[3] = m[3].to(torch.bfloat16)
    arena = FlatParamArena(m)
    dtypes = {g.dtype for g in arena.groups}
    assert dtypes == {torch.float32, torch.bfloat16}
    assert arena.check_views()
    assert arena.numel == sum(p.numel() for p in m.parameters())


def test_float_buffers_reparented_int_buffers_untouched():
    m = make_model()
    m.train()
    m(torch.randn(16, 8))
    arena = FlatParamArena(m)
    assert len(arena.buffer_groups) == 1
    bg = arena.buffer_groups[0]
    bn = m[2]
    lo, hi = bg.slices["2.running_mean"]
    assert bn.running_mean.data_ptr() == bg.flat[lo:hi].data_ptr()
    assert "2.num_batches_tracked" not in bg.slices
    sd = m.state_dict()
    assert torch.equal(sd["2.running_mean"], bn.running_mean)


def test_training_step_keeps_views():
    m = make_model()
    arena = FlatParamArena(m)
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    for _ in range(3):
        opt.zero_grad()
        m(torch.randn(8, 8)).pow(2).mean().backward()
        opt.step()
    assert arena.check_views()


def test_arena_optimizer_matches_per_param():
    """FusedSGD.from_arena (one kernel per group) must match per-parameter
    FusedSGD exactly on CPU."""
    from baton_amd.ops.optim import FusedSGD

    torch.manual_seed(1)
    m1, m2 = make_model(), make_model()
    m2.load_state_dict(m1.state_dict())
    arena = FlatParamArena(m1)
    o1 = FusedSGD.from_arena(arena, lr=0.05, momentum=0.9)
    o2 = FusedSGD(m2.parameters(), lr=0.05, momentum=0.9)
    for _ in range(4):
        x = torch.randn(8, 8)
        o1.zero_grad(); o2.zero_grad()
        m1(x).pow(2).mean().backward()
        m2(x).pow(2).mean().backward()
        o1.step(); o2.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6)


def test_resnet_arena_groups():
    from baton_amd.models.resnet import resnet18

    m = resnet18().to(torch.bfloat16)
    arena = FlatParamArena(m)
    dtypes = sorted(str(g.dtype) for g in arena.groups)
    assert dtypes == ["torch.bfloat16", "torch.float32"]
    # conv/linear weights bf16; BN gamma/beta + Linear bias fp32
    assert arena.check_views()
    bf16 = next(g for g in arena.groups if g.dtype == torch.bfloat16)
    assert bf16.flat.numel() > 10_000_000  # ~11M conv/linear weights
