"""Probe: can RCCL run a 2-rank world on ONE MI355X (both ranks cuda:0)?

Run on a GPU box:  python scripts/probe_nccl_2rank_1gpu.py
Prints PROBE-OK / PROBE-REFUSED(+error) — used to decide whether the
1-GPU nccl multi-rank data-plane test is runnable in this pool.
"""

import os
import sys

import torch
import torch.multiprocessing as mp


def _worker(rank, world, port, q):
    try:
        os.environ.update(
            MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
            RANK=str(rank), WORLD_SIZE=str(world),
        )
        import torch.distributed as dist

        torch.cuda.set_device(0)
        dist.init_process_group("nccl")
        t = torch.full((1024,), float(rank + 1), device="cuda:0")
        dist.all_reduce(t)
        torch.cuda.synchronize()
        ok = float(t[0].item()) == 3.0
        dist.destroy_process_group()
        q.put((rank, "ok" if ok else f"bad value {t[0].item()}"))
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"{type(e).__name__}: {e}"))


def main():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29977, q)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
        if p.is_alive():
            p.terminate()
    results = {}
    while not q.empty():
        r, msg = q.get()
        results[r] = msg
    if results.get(0) == "ok" and results.get(1) == "ok":
        print("PROBE-OK: nccl 2 ranks on 1 GPU works")
    else:
        print(f"PROBE-REFUSED: {results}")
    sys.exit(0)


if __name__ == "__main__":
    main()
