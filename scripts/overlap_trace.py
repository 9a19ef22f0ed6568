"""Produce rocprof evidence of the side-stream aggregation overlap.

Runs (nccl, world 1) a loop of: local step kernels -> fedavg_arena
(async_handle=True) -> next step's param-independent kernels -> fence.
Under `rocprofv3 --kernel-trace` the dispatch table then shows the
aggregation's scale_cast / cast_copy kernels on a DIFFERENT stream with
timestamps overlapping the compute stream's next-step kernels. At world
size 1 the RCCL calls are self-copies (2-rank worlds on one GPU are
refused — profiles/r02_nccl_2rank_1gpu_probe.log), so this demonstrates
the overlap STRUCTURE; the link time it hides appears only on the
driver's multi-GPU runs.

    rocprofv3 --kernel-trace -d OUT -- python scripts/overlap_trace.py
    python benchmarks/overlap_report.py OUT
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29915",
                  RANK="0", WORLD_SIZE="1")

from baton_amd.parallel.data_plane import FederatedDataPlane
from baton_amd.runtime.arena import FlatParamArena
from baton_amd.utils.config import DataPlaneConfig


def main():
    torch.manual_seed(0)
    plane = FederatedDataPlane(DataPlaneConfig(backend="nccl"),
                               device=torch.device("cuda", 0))
    model = torch.nn.Sequential(
        torch.nn.Linear(8192, 8192), torch.nn.ReLU(),
        torch.nn.Linear(8192, 8192),
    ).to("cuda").bfloat16()
    arena = FlatParamArena(model)
    x = torch.randn(8192, 8192, device="cuda").bfloat16()
    big_a = torch.randn(8192, 8192, device="cuda").bfloat16()

    for _ in range(6):
        y = model(x[:2048])
        loss = y.float().square().mean()
        loss.backward()
        # aggregation launches on the side stream; the big matmul below is
        # param-independent compute enqueued on the main stream right
        # after — under the trace it co-runs with the aggregation kernels
        plane.fedavg_arena(arena, 128, async_handle=True)
        _ = big_a @ big_a.t()
        for p in model.parameters():
            if p.grad is not None:
                p.grad.zero_()
        plane.pending.wait()           # fence before the next param read
    torch.cuda.synchronize()
    plane.shutdown()
    print("OVERLAP-TRACE-DONE")


if __name__ == "__main__":
    main()
