#!/usr/bin/env python
"""Flagship federated benchmark — the driver contract.

Measures BASELINE.json's headline metric: federated rounds/sec + aggregate
local samples/sec for ResNet-18 N-client FedAvg (E=1 local epoch per round)
on synthetic CIFAR-shaped data with random-init weights, one rank per GPU
over RCCL. Weak scaling: per-GPU (per-client) work is fixed as N grows.

    python bench.py --gpus N --steps K --warmup W
    # N>1 via: python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
    #   --master-addr 127.0.0.1 --master-port P bench.py --gpus N ...

One federated round (= one "step") per client: E=1 epoch of local SGD over
its private shard (HIP kernels: implicit-GEMM conv, fused BN+ReLU, MFMA
linear, fused CE, fused SGD on the flat arena), then FedAvg aggregation
(pre-scaled RCCL reduce to rank 0 + broadcast over xGMI).

Rank 0 prints exactly one JSON line with the whole-job aggregate
samples/sec (sum over all clients).
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--local-samples", type=int, default=4096,
                   help="per-client samples per round (weak scaling)")
    p.add_argument("--batch-size", type=int, default=256)
    p.add_argument("--model", default="resnet18", choices=["resnet18", "resnet50"])
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--epochs-per-round", type=int, default=1)
    p.add_argument("--hip-graph", action="store_true", default=True,
                   help="capture the minibatch step in a hipGraph (default on)")
    p.add_argument("--no-hip-graph", dest="hip_graph", action="store_false")
    return p.parse_args()


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    distributed = world > 1

    on_gpu = torch.cuda.is_available()
    if on_gpu:
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")
    dtype = torch.bfloat16 if (args.dtype == "bf16" and on_gpu) else torch.float32

    plane = None
    if distributed:
        from baton_amd.parallel.data_plane import FederatedDataPlane
        from baton_amd.utils.config import DataPlaneConfig

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        plane = FederatedDataPlane(
            DataPlaneConfig(backend="nccl" if on_gpu else "gloo"), device=device
        )

    from baton_amd.models.resnet import make_synthetic_cifar, resnet18, resnet50
    from baton_amd.ops import functional as BF
    from baton_amd.ops.optim import FusedSGD
    from baton_amd.runtime.arena import FlatParamArena

    torch.manual_seed(1234)  # identical global init on every client
    model_fn = resnet18 if args.model == "resnet18" else resnet50
    model = model_fn(num_classes=10).to(device).to(dtype)
    model.train()
    arena = FlatParamArena(model)
    opt = FusedSGD.from_arena(arena, lr=0.05, momentum=0.9)

    n_local = args.local_samples
    x, y = make_synthetic_cifar(n_local, seed=1000 + rank, dtype=dtype)
    x, y = x.to(device), y.to(device)
    bs = args.batch_size

    loss_fn = lambda logits, t: BF.cross_entropy(logits.contiguous(), t)
    graph_step = None
    if args.hip_graph and on_gpu:
        from baton_amd.runtime.graph import GraphedTrainStep

        graph_step = GraphedTrainStep(model, opt, loss_fn, x[:bs], y[:bs])

    def one_round():
        if graph_step is not None:
            for e in range(args.epochs_per_round):
                for i in range(0, n_local, bs):
                    loss = graph_step(x[i : i + bs], y[i : i + bs])
        else:
            for e in range(args.epochs_per_round):
                for i in range(0, n_local, bs):
                    bx, by = x[i : i + bs], y[i : i + bs]
                    opt.zero_grad()
                    loss = loss_fn(model(bx), by)
                    loss.backward()
                    opt.step()
        if plane is not None:
            plane.fedavg_arena(arena, n_local)
        return loss

    def sync():
        if plane is not None:
            plane.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        one_round()
    sync()
    t0 = time.perf_counter()
    last_loss = None
    for _ in range(args.steps):
        last_loss = one_round()
    sync()
    elapsed = time.perf_counter() - t0

    # MAX elapsed over ranks
    if plane is not None:
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=plane._coll_device())
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    rounds_per_sec = args.steps / elapsed
    samples_per_sec = world * n_local * args.epochs_per_round * args.steps / elapsed
    if rank == 0:
        result = {
            "metric": "fed_local_samples_per_sec",
            "value": samples_per_sec,
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if dtype == torch.bfloat16 else "fp32",
            "data": "synthetic (CIFAR-shaped NHWC, random labels, random-init weights)",
            "config": {
                "model": args.model,
                "global_batch": args.batch_size * world,
                "seq_len": None,
                "image": "32x32x3",
                "parallelism": f"federated-dp{world} (FedAvg E={args.epochs_per_round})",
                "local_samples_per_round": n_local,
                "rounds_per_sec": rounds_per_sec,
                "last_loss": float(last_loss.item()) if last_loss is not None else None,
            },
        }
        print(json.dumps(result))

    if plane is not None:
        plane.shutdown()


if __name__ == "__main__":
    main()
