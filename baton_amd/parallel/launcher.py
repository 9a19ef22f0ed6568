"""One-command launcher for the full federated deployment on one node:
an HTTP manager + N GPU-client processes (one per GPU) with the RCCL data
plane.

    python -m baton_amd.parallel.launcher --model resnet18 --clients 8 \
        --rounds 5 --n-epoch 1 --local-samples 4096

The parent process runs the manager (aiohttp, Baton wire API) and spawns N
client subprocesses; each client pins its GPU (LOCAL_RANK), builds the
model + FlatParamArena + FederatedDataPlane (RCCL over xGMI; gloo on CPU),
registers over HTTP and serves rounds: local training with the HIP kernel
stack, then pre-scaled reduce + broadcast. The parent triggers the
requested number of rounds and prints the metrics/loss history.
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import subprocess
import sys
import time

import torch


def client_main(args) -> None:
    """Entry for a spawned client (RANK/WORLD_SIZE/MASTER_* in env)."""
    from aiohttp import web

    from baton_amd.models.factory import create_model, make_data
    from baton_amd.parallel.data_plane import FederatedDataPlane
    from baton_amd.parallel.gpu_worker import GPUExperimentWorker
    from baton_amd.runtime.arena import FlatParamArena
    from baton_amd.utils.config import BatonConfig, DataPlaneConfig, TrainConfig

    rank = int(os.environ["RANK"])
    on_gpu = torch.cuda.is_available()
    dtype = torch.bfloat16 if (args.dtype == "bf16" and on_gpu) else torch.float32

    cfg = BatonConfig()
    cfg.train = TrainConfig(
        n_epoch=args.n_epoch,
        lr=args.lr,
        batch_size=args.batch_size,
        optimizer=args.optimizer,
        momentum=0.9 if args.optimizer == "sgd" else 0.0,
        fedprox_mu=args.fedprox_mu,
    )
    cfg.control.heartbeat_interval = 5.0

    plane = FederatedDataPlane(
        DataPlaneConfig(backend="nccl" if on_gpu else "gloo"),
    )
    torch.manual_seed(args.seed)  # identical global init on every client
    model = create_model(args.model, cfg.train).to(plane.device).to(dtype)
    arena = FlatParamArena(model)

    class Client(GPUExperimentWorker):
        def get_data(self):
            data, n = make_data(
                args.model, args.local_samples, seed=args.seed + 1000 + rank,
                seq_len=args.seq_len, dtype=dtype,
            )
            return tuple(t.to(plane.device) for t in data), n

    async def serve():
        app = web.Application(client_max_size=1 << 30)
        worker = Client(
            app, model,
            manager_url=f"http://127.0.0.1:{args.port}",
            plane=plane, arena=arena,
            port=args.port + 1 + rank,
            name=args.model,
            config=cfg, auto_register=False,
        )
        runner = web.AppRunner(app)
        await runner.setup()
        site = web.TCPSite(runner, "127.0.0.1", args.port + 1 + rank)
        await site.start()
        await worker.register_with_manager()
        while not os.path.exists(args.done_file):
            await asyncio.sleep(0.2)
        # drain a round that may still be running
        while worker._busy.locked():
            await asyncio.sleep(0.2)
        plane.barrier()
        await worker.stop()
        await runner.cleanup()

    asyncio.run(serve())
    plane.shutdown()


async def manager_main(args) -> dict:
    from aiohttp import web

    from baton_amd.control.manager import Manager
    from baton_amd.models.factory import create_model
    from baton_amd.utils.config import BatonConfig, TrainConfig

    cfg = BatonConfig()
    cfg.train = TrainConfig(n_epoch=args.n_epoch, batch_size=args.batch_size)
    cfg.control.aggregation_mode = "rccl"  # this launcher IS the rccl topology
    cfg.checkpoint_dir = args.checkpoint_dir
    app = web.Application(client_max_size=1 << 30)
    manager = Manager(app, config=cfg)
    torch.manual_seed(args.seed)
    gmodel = create_model(args.model, cfg.train)
    if args.dtype == "bf16" and torch.cuda.is_available():
        gmodel = gmodel.to(torch.bfloat16)
    exp = manager.register_experiment(gmodel, name=args.model)
    if args.resume:
        exp.load_checkpoint()
    runner = web.AppRunner(app)
    await runner.setup()
    site = web.TCPSite(runner, "127.0.0.1", args.port)
    await site.start()

    # wait for all clients
    deadline = time.time() + 300
    while len(exp.registry) < args.clients and time.time() < deadline:
        await asyncio.sleep(0.2)
    assert len(exp.registry) == args.clients, (
        f"only {len(exp.registry)}/{args.clients} clients registered"
    )
    print(f"[launcher] {args.clients} clients registered; running "
          f"{args.rounds} rounds of E={args.n_epoch}")

    t0 = time.perf_counter()
    for r in range(args.rounds):
        assert await exp.start_round(n_epoch=args.n_epoch)
        while exp.rounds.in_progress:
            await asyncio.sleep(0.05)
        print(f"[launcher] round {r}: {exp.rounds.round_log[-1]}")
    elapsed = time.perf_counter() - t0

    result = {
        "rounds": args.rounds,
        "elapsed_sec": elapsed,
        "rounds_per_sec": args.rounds / elapsed,
        "samples_per_sec": args.clients * args.local_samples * args.n_epoch
        * args.rounds / elapsed,
        "loss_history": exp.rounds.loss_history,
        "metrics": exp.metrics,
    }
    with open(args.done_file, "w") as f:
        f.write("done")
    await asyncio.sleep(1.0)
    await manager.stop()
    await runner.cleanup()
    return result


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet18")
    p.add_argument("--clients", type=int, default=torch.cuda.device_count() or 2)
    p.add_argument("--rounds", type=int, default=5)
    p.add_argument("--n-epoch", type=int, default=1)
    p.add_argument("--local-samples", type=int, default=4096)
    p.add_argument("--batch-size", type=int, default=256)
    p.add_argument("--seq-len", type=int, default=128)
    p.add_argument("--lr", type=float, default=0.05)
    p.add_argument("--optimizer", default="sgd", choices=["sgd", "adam"])
    p.add_argument("--fedprox-mu", type=float, default=0.0)
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--port", type=int, default=8080)
    p.add_argument("--dist-port", type=int, default=29531)
    p.add_argument("--seed", type=int, default=1234)
    p.add_argument("--checkpoint-dir", default=None)
    p.add_argument("--resume", action="store_true")
    p.add_argument("--done-file", default="/tmp/baton_launcher_done")
    p.add_argument("--_client", action="store_true", help=argparse.SUPPRESS)
    args = p.parse_args()

    if args._client:
        client_main(args)
        return

    if os.path.exists(args.done_file):
        os.remove(args.done_file)

    procs = []
    for rank in range(args.clients):
        env = dict(os.environ)
        env.update(
            RANK=str(rank),
            LOCAL_RANK=str(rank),
            WORLD_SIZE=str(args.clients),
            MASTER_ADDR="127.0.0.1",
            MASTER_PORT=str(args.dist_port),
        )
        procs.append(
            subprocess.Popen(
                [sys.executable, "-m", "baton_amd.parallel.launcher", "--_client"]
                + sys.argv[1:],
                env=env,
            )
        )
    try:
        result = asyncio.run(manager_main(args))
        print(json.dumps({k: v for k, v in result.items() if k != "loss_history"}))
        print(f"[launcher] final losses: {result['loss_history'][-5:]}")
    finally:
        for pr in procs:
            try:
                pr.wait(timeout=60)
            except subprocess.TimeoutExpired:
                pr.terminate()


if __name__ == "__main__":
    main()
