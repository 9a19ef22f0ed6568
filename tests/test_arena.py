"""FlatParamArena invariants (runtime/arena.py)."""

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
    assert arena.check_views()
    # values preserved
    for n, p in m.named_parameters():
        assert torch.equal(p.detach(), before[n])
    # mutating flat mutates the params
    arena.flat_params.zero_()
    for p in m.parameters():
        assert p.detach().abs().sum() == 0


def test_backward_accumulates_into_flat_grads():
    m = make_model()
    arena = FlatParamArena(m)
    x = torch.randn(4, 8)
    m(x).sum().backward()
    g = arena.flat_grads
    assert g.abs().sum() > 0
    for n, p in m.named_parameters():
        lo, hi = arena.param_slices[n]
        assert p.grad.data_ptr() == g[lo:hi].data_ptr()
    arena.zero_grads()
    assert g.abs().sum() == 0


def test_float_buffers_reparented_int_buffers_untouched():
    m = make_model()
    m.train()
    m(torch.randn(16, 8))  # populate BN running stats
    arena = FlatParamArena(m)
    assert arena.flat_buffers is not None
    bn = m[2]
    # running_mean is a view into flat_buffers
    lo, hi = arena.buffer_slices["2.running_mean"]
    assert bn.running_mean.data_ptr() == arena.flat_buffers[lo:hi].data_ptr()
    # int counter not in the flat buffer
    assert "2.num_batches_tracked" not in arena.buffer_slices
    # state_dict still reflects the views
    sd = m.state_dict()
    assert torch.equal(sd["2.running_mean"], bn.running_mean)


def test_training_step_keeps_views():
    m = make_model()
    arena = FlatParamArena(m)
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    for _ in range(3):
        opt.zero_grad()
        loss = m(torch.randn(8, 8)).pow(2).mean()
        loss.backward()
        opt.step()
    assert arena.check_views()


def test_load_flat_roundtrip():
    m = make_model()
    arena = FlatParamArena(m)
    new = torch.randn_like(arena.flat_params)
    arena.load_flat(new)
    assert torch.equal(arena.flat_params, new)
    # params see the new values
    n0, p0 = next(iter(m.named_parameters()))
    lo, hi = arena.param_slices[n0]
    assert torch.equal(p0.detach().reshape(-1), new[lo:hi])


def test_mixed_dtype_rejected():
    m = make_model()
    m[0] = m[0].to(torch.bfloat16)
    with pytest.raises(ValueError):
        FlatParamArena(m)
