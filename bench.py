#!/usr/bin/env python
"""Flagship federated benchmark — the driver contract.

Measures BASELINE.json's headline metric: federated rounds/sec + aggregate
local samples/sec for ResNet-18 N-client FedAvg (E=1 local epoch per round)
on synthetic CIFAR-shaped data with random-init weights, one rank per GPU
over RCCL. Weak scaling: per-GPU (per-client) work is fixed as N grows.

    python bench.py --gpus N --steps K --warmup W
    # N>1 via: python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
    #   --master-addr 127.0.0.1 --master-port P bench.py --gpus N ...

One federated round (= one "step") per client: E local epochs of training
over its private shard (HIP kernels: implicit-GEMM conv / MFMA GEMMs /
fused norms / fused losses / fused SGD-Adam on the flat arena; the
minibatch step is hipGraph-captured), then FedAvg aggregation (pre-scaled
RCCL reduce to rank 0 + broadcast over xGMI).

Other BASELINE configs run through the same harness:
    --model bert-base --optimizer adam --epochs-per-round 5   (config 3)
    --model llama-lora --seq-len 512 --optimizer adam         (config 4,
        adapter-delta-only reduce: only the LoRA arena crosses xGMI)
    --model resnet50 --fedprox-mu 0.01                        (config 5)

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
    p.add_argument("--local-samples", type=int, default=0,
                   help="per-client samples per round (weak scaling); "
                        "0 = per-model default")
    p.add_argument("--batch-size", type=int, default=0,
                   help="0 = per-model default")
    p.add_argument("--model", default="resnet18",
                   choices=["resnet18", "resnet50", "bert-base", "bert-tiny",
                            "llama-lora", "llama-tiny"])
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--epochs-per-round", type=int, default=1)
    p.add_argument("--seq-len", type=int, default=0,
                   help="0 = per-model default")
    p.add_argument("--optimizer", default="", choices=["", "sgd", "adam"])
    p.add_argument("--fedprox-mu", type=float, default=0.0)
    p.add_argument("--dirichlet-alpha", type=float, default=0.0,
                   help="> 0: non-IID label split of a shared synthetic set "
                        "across clients (config 5); 0 = IID per-client data")
    p.add_argument("--lr", type=float, default=0.0, help="0 = default")
    p.add_argument("--hip-graph", dest="hip_graph", action="store_true",
                   default=None,
                   help="capture the WHOLE local round (all minibatches) as "
                        "one hipGraph, replayed per round (default: on for "
                        "graph-capturable models; single-minibatch capture "
                        "measured slower than eager and was replaced)")
    p.add_argument("--no-hip-graph", dest="hip_graph", action="store_false")
    return p.parse_args()


MODEL_DEFAULTS = {
    # batch, optimizer, lr, graph-capturable, local samples/round, seq len
    # (batch sizes picked from the measured throughput saturation curve)
    # Adam models are NOT graph-capturable: the fused adam kernel takes the
    # host-computed bias corrections 1-beta^t as scalar args (ops/optim.py),
    # which a graph replay would freeze at the captured step count.
    "resnet18": (8192, "sgd", 0.05, True, 16384, 0),
    "resnet50": (4096, "sgd", 0.05, True, 8192, 0),
    "bert-base": (512, "adam", 5e-5, False, 4096, 128),
    "bert-tiny": (32, "adam", 1e-4, False, 256, 64),
    "llama-lora": (32, "adam", 1e-4, False, 64, 512),
    "llama-tiny": (8, "adam", 1e-4, False, 64, 64),
}


def main():
    args = parse_args()
    # Self-launch: `python bench.py --gpus N` with no torchrun rendezvous in
    # the environment spawns its own N ranks (driver contract: one command
    # must produce an honest N-GPU record).
    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        import subprocess
        import sys

        cmd = [
            sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
            f"--nproc-per-node={args.gpus}", "--master-addr=127.0.0.1",
            f"--master-port={os.environ.get('BENCH_MASTER_PORT', '29531')}",
            os.path.abspath(__file__), *sys.argv[1:],
        ]
        raise SystemExit(subprocess.run(cmd).returncode)

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    distributed = world > 1
    if args.gpus != world:
        raise SystemExit(
            f"bench.py: --gpus {args.gpus} but WORLD_SIZE={world}: refusing "
            f"to report an n_gpus that does not match the actual world"
        )

    d_bs, d_opt, d_lr, graphable, d_local, d_seq = MODEL_DEFAULTS[args.model]
    bs = args.batch_size or d_bs
    optimizer = args.optimizer or d_opt
    lr = args.lr or d_lr
    if args.local_samples == 0:
        args.local_samples = d_local
    if args.seq_len == 0:
        args.seq_len = d_seq

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

    from baton_amd.models.factory import create_model, make_data
    from baton_amd.ops import functional as BF
    from baton_amd.ops.optim import FusedAdam, FusedSGD
    from baton_amd.runtime.arena import FlatParamArena
    from baton_amd.utils.config import TrainConfig

    torch.manual_seed(1234)  # identical global init on every client
    tc = TrainConfig(optimizer=optimizer, lr=lr, batch_size=bs,
                     fedprox_mu=args.fedprox_mu)
    model = create_model(args.model, tc).to(device).to(dtype)
    if args.model.startswith("llama"):
        model.rope_cos = model.rope_cos.float()
        model.rope_sin = model.rope_sin.float()
    model.train()
    is_llama = args.model.startswith("llama")
    arena = FlatParamArena(model, include_buffers=not is_llama)
    if optimizer == "sgd":
        opt = FusedSGD.from_arena(arena, lr=lr, momentum=0.9)
    else:
        opt = FusedAdam.from_arena(arena, lr=lr)

    n_local = args.local_samples
    if args.dirichlet_alpha > 0 and args.model.startswith("resnet"):
        # config 5: one shared synthetic dataset, Dirichlet(alpha) label
        # split; every rank derives the same partition (same seed) and
        # keeps its shard resident in HBM
        from baton_amd.fed.dataset import FederatedTensorDataset

        full, _ = make_data(args.model, n_local * world, seed=999,
                            seq_len=args.seq_len, dtype=dtype)
        ds = FederatedTensorDataset(full, n_clients=world, split="dirichlet",
                                    alpha=args.dirichlet_alpha, label_index=1,
                                    seed=7)
        data = ds.shard(rank, device=device)
        n_local = data[0].shape[0]
        # non-IID shards are uneven: keep at least one batch per client
        bs = min(bs, max(n_local, 1))
    else:
        data, _ = make_data(args.model, n_local, seed=1000 + rank,
                            seq_len=args.seq_len, dtype=dtype)
        data = tuple(t.to(device) for t in data)
    *inputs, target = data

    # model-family loss: resnet -> CE(logits); bert/llama -> model loss head
    if args.model.startswith("resnet"):
        loss_fn = lambda out, t: BF.cross_entropy(out.contiguous(), t)
    elif args.model.startswith("bert"):
        loss_fn = model.mlm_loss
    else:
        loss_fn = model.lm_loss

    # FedProx snapshot (config 5)
    global_params = None
    if args.fedprox_mu > 0:
        global_params = [p.detach().clone() for p in model.parameters()
                        if p.requires_grad]

    def eager_batch(i, fence_pending=False):
        bx = [t[i : i + bs] for t in inputs]
        by = target[i : i + bs]
        opt.zero_grad()
        if fence_pending and plane is not None and plane.pending is not None:
            # overlap: grad zero-fill + input slicing above ran while the
            # side-stream collectives were still in flight; fence the
            # compute stream before the first read of the averaged params
            plane.pending.wait()
        loss = loss_fn(model(*bx), by)
        loss.backward()
        if global_params is not None:
            with torch.no_grad():
                for p, g in zip(
                    (p for p in model.parameters() if p.requires_grad),
                    global_params,
                ):
                    if p.grad is not None:
                        p.grad.add_(p.detach() - g, alpha=args.fedprox_mu)
        opt.step()
        return loss

    # FedProx must snapshot the averaged params right after aggregation, so
    # it takes the fenced (synchronous) path; plain FedAvg overlaps the
    # collectives with the next round's param-independent prep.
    use_async = global_params is None

    # hipGraph-captured aggregation (the manager-side round sequence on the
    # data plane) — opt-in until RCCL graph capture is validated at N>1
    if (plane is not None and on_gpu and
            os.environ.get("BATON_GRAPH_AGG", "0") == "1"):
        try:
            plane.capture_aggregation(arena, n_local)
        except Exception as e:  # noqa: BLE001
            print(f"# aggregation graph capture unavailable: {e}")

    # whole-round hipGraph capture: the round iterates fixed slices of
    # HBM-resident data, so the full epochs-x-batches loop captures as ONE
    # graph with zero per-replay copies (north-star "per-round worker step
    # is hipGraph-captured"). Default-on where capturable.
    use_graph = args.hip_graph if args.hip_graph is not None else graphable
    graph_round = None
    if use_graph and on_gpu and graphable and args.fedprox_mu == 0:
        from baton_amd.runtime.graph import GraphedRound

        def _capture_round():
            loss = None
            for _ in range(args.epochs_per_round):
                for i in range(0, n_local - bs + 1, bs):
                    loss = eager_batch(i, fence_pending=False)
            return loss

        graph_round = GraphedRound(_capture_round)


    def one_round():
        loss = None
        if graph_round is not None:
            if plane is not None and plane.pending is not None:
                plane.pending.wait()
            loss = graph_round()
        else:
            first = True
            for _ in range(args.epochs_per_round):
                for i in range(0, n_local - bs + 1, bs):
                    loss = eager_batch(i, fence_pending=first)
                    first = False
        if plane is not None:
            plane.fedavg_arena(arena, n_local, async_handle=use_async)
            if global_params is not None:
                for gp, p in zip(global_params,
                                 (p for p in model.parameters() if p.requires_grad)):
                    gp.copy_(p.detach())
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

    samples_per_round = (n_local // bs) * bs  # tail batch dropped
    rounds_per_sec = args.steps / elapsed
    samples_per_sec = (world * samples_per_round * args.epochs_per_round *
                       args.steps / elapsed)
    if rank == 0:
        is_image = args.model.startswith("resnet")
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
            "data": "synthetic (random-init weights; no network for datasets)",
            "config": {
                "model": args.model,
                "global_batch": bs * world,
                "seq_len": None if is_image else args.seq_len,
                "image": "32x32x3" if is_image else None,
                "parallelism": f"federated-dp{world} (FedAvg E={args.epochs_per_round}"
                               + (f", FedProx mu={args.fedprox_mu}" if args.fedprox_mu else "")
                               + (f", Dirichlet a={args.dirichlet_alpha}" if args.dirichlet_alpha else "")
                               + ")",
                "local_samples_per_round": samples_per_round,
                "optimizer": optimizer,
                "hip_graph": graph_round is not None,
                "rounds_per_sec": rounds_per_sec,
                "last_loss": float(last_loss.item()) if last_loss is not None else None,
            },
        }
        print(json.dumps(result))

    if plane is not None:
        plane.shutdown()


if __name__ == "__main__":
    main()
