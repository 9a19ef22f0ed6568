"""Multi-process data-plane tests: gloo, world_size=2, on CPU.

Asserts the RCCL/gloo pre-scaled reduce+broadcast path produces the SAME
global model as the CPU FedAvg oracle (fed.aggregate.fedavg_) — the
equivalence that keeps the HTTP path and the GPU fast path honest
(SURVEY.md §7 step 3).
"""

from __future__ import annotations

import os
from collections import OrderedDict

import pytest
import torch
import torch.multiprocessing as mp
import torch.nn as nn

from baton_amd.fed.aggregate import fedavg_


def _free_port() -> int:
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def build_client_model(rank: int) -> nn.Module:
    """Deterministic per-rank model incl. BatchNorm (int + float buffers)."""
    torch.manual_seed(100 + rank)
    m = nn.Sequential(nn.Linear(6, 8), nn.BatchNorm1d(8), nn.Linear(8, 2))
    m.train()
    for _ in range(rank + 1):  # distinct num_batches_tracked per rank
        m(torch.randn(8, 6))
    return m


N_SAMPLES = {0: 96, 1: 288}  # rank 1 is the heaviest client


def _oracle_state_dict(world: int) -> "OrderedDict[str, torch.Tensor]":
    global_m = build_client_model(0)
    sds = [OrderedDict(build_client_model(r).state_dict()) for r in range(world)]
    weights = [N_SAMPLES[r] for r in range(world)]
    fedavg_(global_m.state_dict(), sds, weights)
    return OrderedDict(
        (k, v.detach().clone()) for k, v in global_m.state_dict().items()
    )


def _worker(rank: int, world: int, port: int, mode: str, out_dir: str):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    from baton_amd.parallel.data_plane import FederatedDataPlane
    from baton_amd.runtime.arena import FlatParamArena
    from baton_amd.utils.config import DataPlaneConfig

    cfg = DataPlaneConfig(backend="gloo", master_port=port)
    plane = FederatedDataPlane(cfg, device=torch.device("cpu"))
    model = build_client_model(rank)

    if mode == "arena":
        arena = FlatParamArena(model)
        weights = plane.fedavg_arena(arena, N_SAMPLES[rank])
    else:
        weights = plane.fedavg_model(model, N_SAMPLES[rank])

    assert weights.tolist() == [96.0, 288.0]

    # weighted loss mixing
    losses = plane.weighted_mean_losses([1.0 * (rank + 1)], weights)
    expect = (1.0 * 96 + 2.0 * 288) / 384
    assert abs(losses[0] - expect) < 1e-9, losses

    torch.save(
        OrderedDict((k, v.detach().clone()) for k, v in model.state_dict().items()),
        os.path.join(out_dir, f"rank{rank}.pt"),
    )
    plane.barrier()
    plane.shutdown()


@pytest.mark.parametrize("mode", ["arena", "model"])
def test_gloo_fedavg_matches_cpu_oracle(tmp_path, mode):
    world = 2
    port = _free_port()
    ctx = mp.get_context("spawn")
    procs = [
        ctx.Process(target=_worker, args=(r, world, port, mode, str(tmp_path)))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0, f"worker failed with {p.exitcode}"

    oracle = _oracle_state_dict(world)
    for r in range(world):
        got = torch.load(tmp_path / f"rank{r}.pt", weights_only=True)
        assert list(got) == list(oracle)
        for k in oracle:
            if oracle[k].is_floating_point():
                assert torch.allclose(got[k], oracle[k], rtol=1e-6, atol=1e-7), (
                    f"rank {r} key {k} mismatch"
                )
            else:
                assert torch.equal(got[k], oracle[k]), f"rank {r} key {k}"
    # every rank converged to the SAME global model
    g0 = torch.load(tmp_path / "rank0.pt", weights_only=True)
    g1 = torch.load(tmp_path / "rank1.pt", weights_only=True)
    for k in g0:
        assert torch.equal(g0[k], g1[k])
