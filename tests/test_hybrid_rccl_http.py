"""Full hybrid topology test, CPU/gloo, world_size=2, real processes:
HTTP control plane (manager + 2 workers, reference wire API) + collective
data plane (FederatedDataPlane.fedavg_arena) — the GPU deployment shape of
parallel/gpu_worker.py exercised without a GPU (the
HTTP path kept as cross-check oracle)."""

from __future__ import annotations

import asyncio
import os
import socket

import pytest
import torch
import torch.multiprocessing as mp


def _free_ports(n):
    socks = [socket.socket() for _ in range(n)]
    for s in socks:
        s.bind(("127.0.0.1", 0))
    ports = [s.getsockname()[1] for s in socks]
    for s in socks:
        s.close()
    return ports


def _worker_proc(rank, world, dist_port, manager_port, worker_ports, out_dir):
    os.environ.update(
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT=str(dist_port),
        RANK=str(rank),
        WORLD_SIZE=str(world),
    )
    from aiohttp import web

    from baton_amd.control.manager import Manager
    from baton_amd.models.mlp import LinearRegressionModel, make_synthetic_regression
    from baton_amd.parallel.data_plane import FederatedDataPlane
    from baton_amd.parallel.gpu_worker import GPUExperimentWorker
    from baton_amd.runtime.arena import FlatParamArena
    from baton_amd.utils.config import BatonConfig, DataPlaneConfig

    cfg = BatonConfig()
    cfg.control.heartbeat_interval = 0.5
    cfg.control.aggregation_mode = "rccl"  # server-side switch
    cfg.train.n_epoch = 2
    cfg.train.lr = 0.02

    plane = FederatedDataPlane(
        DataPlaneConfig(backend="gloo", master_port=dist_port),
        device=torch.device("cpu"),
    )

    torch.manual_seed(7)  # identical global init on every rank
    model = LinearRegressionModel(cfg.train)
    arena = FlatParamArena(model)

    class Worker(GPUExperimentWorker):
        def get_data(self):
            n = 128 * (rank + 1)  # distinct shard sizes -> real weighting
            x, y = make_synthetic_regression(n, seed=500 + rank)
            return (x, y), n

    async def main():
        apps = []
        wapp = web.Application(client_max_size=1 << 30)
        worker = Worker(
            wapp,
            model,
            manager_url=f"http://127.0.0.1:{manager_port}",
            plane=plane,
            arena=arena,
            port=worker_ports[rank],
            config=cfg,
            auto_register=False,
        )
        runner = web.AppRunner(wapp)
        await runner.setup()
        site = web.TCPSite(runner, "127.0.0.1", worker_ports[rank])
        await site.start()
        apps.append(runner)

        exp = None
        if rank == 0:
            mapp = web.Application(client_max_size=1 << 30)
            manager = Manager(mapp, config=cfg)
            torch.manual_seed(7)
            gmodel = LinearRegressionModel(cfg.train)
            exp = manager.register_experiment(gmodel)
            mrunner = web.AppRunner(mapp)
            await mrunner.setup()
            msite = web.TCPSite(mrunner, "127.0.0.1", manager_port)
            await msite.start()
            apps.append(mrunner)

        await worker.register_with_manager()

        if rank == 0:
            # wait for both clients, run 2 rounds, record results
            for _ in range(400):
                if len(exp.registry) == 2:
                    break
                await asyncio.sleep(0.05)
            assert len(exp.registry) == 2, "both workers must register"
            for _ in range(2):
                assert await exp.start_round(n_epoch=cfg.train.n_epoch)
                for _ in range(600):
                    if not exp.rounds.in_progress:
                        break
                    await asyncio.sleep(0.05)
                assert not exp.rounds.in_progress, "round hung"
            torch.save(
                {
                    "manager_sd": {k: v.clone() for k, v in exp.model.state_dict().items()},
                    "loss_history": exp.rounds.loss_history,
                },
                os.path.join(out_dir, "manager.pt"),
            )
            with open(os.path.join(out_dir, "rounds_done"), "w") as f:
                f.write("ok")
        else:
            # keep serving rounds until rank 0 reports completion — entering
            # the barrier early would interleave with the round collectives
            for _ in range(1200):
                if os.path.exists(os.path.join(out_dir, "rounds_done")) and \
                        not worker._busy.locked():
                    break
                await asyncio.sleep(0.05)
        # every rank saves its local (post-broadcast) model
        torch.save(
            {k: v.clone() for k, v in model.state_dict().items()},
            os.path.join(out_dir, f"rank{rank}.pt"),
        )
        # rendezvous so nobody exits before the collectives finish
        plane.barrier()
        await worker.stop()
        for r in apps:
            await r.cleanup()

    asyncio.run(main())
    plane.shutdown()


def test_hybrid_http_control_rccl_data(tmp_path):
    world = 2
    dist_port, manager_port, w0, w1 = _free_ports(4)
    ctx = mp.get_context("spawn")
    procs = [
        ctx.Process(
            target=_worker_proc,
            args=(r, world, dist_port, manager_port, [w0, w1], str(tmp_path)),
        )
        for r in range(world)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0, f"hybrid worker exited {p.exitcode}"

    r0 = torch.load(tmp_path / "rank0.pt", weights_only=True)
    r1 = torch.load(tmp_path / "rank1.pt", weights_only=True)
    mgr = torch.load(tmp_path / "manager.pt", weights_only=False)
    # broadcast left every client with the SAME global model
    for k in r0:
        assert torch.equal(r0[k], r1[k]), f"clients diverged on {k}"
    # manager copied rank 0's aggregated weights (rccl mode)
    for k in r0:
        assert torch.equal(mgr["manager_sd"][k], r0[k]), f"manager stale on {k}"
    # loss history recorded (2 rounds x 2 epochs)
    assert len(mgr["loss_history"]) == 4
    assert mgr["loss_history"][-1] < mgr["loss_history"][0]
