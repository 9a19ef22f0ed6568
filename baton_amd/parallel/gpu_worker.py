"""GPU federated client: the reference worker runtime bound to one MI355X.

This is the production topology of SURVEY.md §1's "layer-map implication":
the aiohttp CONTROL plane (registration, heartbeat, round triggers — wire
format and routes identical to the HTTP-only worker) stays, while the theta
DATA plane moves to RCCL over xGMI: after local training, all N GPU-clients
enter a pre-scaled reduce + broadcast (FederatedDataPlane.fedavg_arena), so
the full parameter traffic never touches TCP or pickle-era serialization.

Protocol in 'rccl' aggregation mode:
  * every worker reports {n_samples, loss_history, update_name} as usual;
  * ONLY rank 0 attaches the (already globally averaged) state_dict, which
    the manager COPIES into its global model (for checkpoints and late
    joiners) instead of re-averaging — Experiment.end_round handles this.

Static membership: the RCCL world is fixed at launch (one process per
GPU). A client that accepts round_start MUST reach the collective, so the
manager should run with partial_policy='abort' + all-accepted rounds in
this mode; elastic membership remains available via the HTTP data path.
"""

from __future__ import annotations

import logging
from typing import List, Optional, Tuple

import torch

from baton_amd.control.worker import ExperimentWorker
from baton_amd.parallel.data_plane import FederatedDataPlane
from baton_amd.runtime.arena import FlatParamArena

log = logging.getLogger("baton.gpu_worker")


class GPUExperimentWorker(ExperimentWorker):
    def __init__(
        self,
        app,
        model: torch.nn.Module,
        manager_url: str,
        plane: FederatedDataPlane,
        arena: Optional[FlatParamArena] = None,
        **kwargs,
    ):
        super().__init__(app, model, manager_url, **kwargs)
        self.plane = plane
        self.arena = arena if arena is not None else FlatParamArena(model)
        self.device = plane.device
        model.to(self.device)

    def _train_locally(self, state_dict, n_epoch: int) -> Tuple[int, List[float]]:
        # install global weights (state dict arrives over HTTP; in steady
        # state it equals what the last broadcast already left in the arena)
        self.model.load_state_dict(state_dict)
        data, n_samples = self.get_data()
        if self.trainer is not None:
            loss_history = self.trainer(self.model, data, n_epoch)
        else:
            loss_history = self.model.train_round(*data, n_epoch=n_epoch)
        # ---- data plane: FedAvg over xGMI (all clients rendezvous here)
        weights = self.plane.fedavg_arena(self.arena, n_samples)
        loss_history = self.plane.weighted_mean_losses(loss_history, weights)
        return n_samples, [float(x) for x in loss_history]

    async def report_update(self, update_name, n_samples, loss_history):
        """Rank 0 attaches the averaged state_dict; other ranks send meta
        only (the manager copies rank 0's weights — no re-averaging)."""
        from baton_amd.control.wire import encode_payload
        import aiohttp

        include_sd = self.plane.rank == 0
        payload = encode_payload(
            {
                "update_name": update_name,
                "n_samples": int(n_samples),
                "loss_history": loss_history,
                "aggregated": True,
            },
            self.model.state_dict() if include_sd else None,
        )
        sess = await self.session()
        try:
            async with sess.post(
                f"{self.manager_url}/{self.experiment}/update",
                params={"client_id": self.client_id, "key": self.key},
                data=payload,
                timeout=aiohttp.ClientTimeout(total=300),
            ) as resp:
                if resp.status not in (200,):
                    log.warning("gpu report got HTTP %d", resp.status)
        except (aiohttp.ClientError, OSError) as e:
            log.warning("gpu report failed: %s", e)
