"""RCCL data-plane smoke on real hardware (1 rank).

A 2-rank world on one MI355X is refused by RCCL ("Duplicate GPU
detected" — profiles/r02_nccl_2rank_1gpu_probe.log), so multi-rank runs
only on the driver's 8-GPU node. This test still executes the ENTIRE
nccl-backend aggregation machinery on hardware — process-group init over
RCCL, the side-stream bucketed pipeline (HIP events, async_op reduce /
broadcast on the communicator stream, fused scale_cast / cast_copy), the
PendingAggregation fence, and the weighted loss all-reduce — with the
world-size-1 identity as the oracle (n/N = 1, reduce+broadcast are
self-copies, so the model must be bit-identical after aggregation).
"""

import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

_SNIPPET = r"""
import os
import torch

os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29913",
                  RANK="0", WORLD_SIZE="1")
from baton_amd.parallel.data_plane import FederatedDataPlane
from baton_amd.runtime.arena import FlatParamArena
from baton_amd.utils.config import DataPlaneConfig

torch.manual_seed(0)
plane = FederatedDataPlane(DataPlaneConfig(backend="nccl"),
                           device=torch.device("cuda", 0))
assert plane._side_stream is not None, "overlap side stream must exist"

model = torch.nn.Sequential(
    torch.nn.Linear(64, 128), torch.nn.BatchNorm1d(128),
    torch.nn.Linear(128, 32),
).to("cuda").bfloat16()
model.train()
model(torch.randn(8, 64, device="cuda").bfloat16())  # bump int buffers
before = {k: v.detach().clone() for k, v in model.state_dict().items()}

arena = FlatParamArena(model)
# async handle path: fence later, exactly like bench.py's overlap
plane.fedavg_arena(arena, 123, async_handle=True)
assert plane.pending is not None
w = plane.pending.wait()
torch.cuda.synchronize()
assert w.tolist() == [123.0]

after = model.state_dict()
for k in before:
    assert torch.equal(before[k], after[k]), f"{k} changed under identity"

losses = plane.weighted_mean_losses([1.0, 2.0], w)
assert losses == [1.0, 2.0]
plane.shutdown()
print("NCCL-1RANK-OK")
"""


def test_nccl_world1_sidestream_identity():
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.run([sys.executable, "-c", _SNIPPET], env=env,
                          capture_output=True, text=True, timeout=600)
    assert proc.returncode == 0, (
        f"nccl world-1 smoke failed:\n{proc.stdout[-2000:]}\n{proc.stderr[-4000:]}"
    )
    assert "NCCL-1RANK-OK" in proc.stdout


_CAPTURE_SNIPPET = r"""
import os
import torch

os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29914",
                  RANK="0", WORLD_SIZE="1")
from baton_amd.parallel.data_plane import FederatedDataPlane
from baton_amd.runtime.arena import FlatParamArena
from baton_amd.utils.config import DataPlaneConfig

torch.manual_seed(0)
plane = FederatedDataPlane(DataPlaneConfig(backend="nccl"),
                           device=torch.device("cuda", 0))
model = torch.nn.Linear(64, 64).to("cuda").bfloat16()
before = {k: v.detach().clone() for k, v in model.state_dict().items()}
arena = FlatParamArena(model)
print("CAPTURE-START", flush=True)
ok = plane.capture_aggregation(arena, 123)
assert ok
for _ in range(2):
    plane.fedavg_arena(arena, 123)
torch.cuda.synchronize()
for k in before:
    assert torch.equal(before[k], model.state_dict()[k])
# changed sample count -> graph dropped, eager fallback still correct
w3 = plane.fedavg_arena(arena, 77)
torch.cuda.synchronize()
assert w3.tolist() == [77.0]
plane.shutdown()
print("CAPTURE-OK")
"""


def test_nccl_world1_graph_captured_aggregation():
    """hipGraph capture of the aggregation sequence (RCCL collectives in a
    graph). Runs in a subprocess because an unsupported RCCL-in-graph
    stack crashes rather than erroring; a crash AFTER the CAPTURE-START
    marker records the capability as unsupported (skip) instead of
    failing the build."""
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.run([sys.executable, "-c", _CAPTURE_SNIPPET], env=env,
                          capture_output=True, text=True, timeout=600)
    if proc.returncode == 0 and "CAPTURE-OK" in proc.stdout:
        return
    if "CAPTURE-START" in proc.stdout:
        pytest.skip(
            f"RCCL graph capture unsupported on this stack "
            f"(rc={proc.returncode}); aggregation graphing stays opt-in"
        )
    raise AssertionError(
        f"capture probe failed before capture:\n{proc.stdout[-2000:]}\n"
        f"{proc.stderr[-4000:]}"
    )
