"""Manager: per-experiment round orchestration + FedAvg aggregation.

Re-designed equivalent of the reference's Manager/Experiment
(/root/reference/manager.py:10-133). HTTP surface preserved (SURVEY.md
§2.4): routes ``POST /{exp}/update``, ``GET /{exp}/start_round``,
``GET /{exp}/end_round``, ``GET /{exp}/loss_history`` (fixed — defect D1),
plus registry routes; status codes 200/401/410/423 kept.

Fixed-by-design relative to the reference:
  D1  loss_history / round-state endpoints return real data;
  D3/D7  rounds have an optional deadline with an explicit
         partial-participation policy instead of hanging forever;
  D4  aggregation via fed.aggregate.fedavg_ (0-dim + integer-buffer safe);
  D6  no pickle on the wire (control/wire.py), crypto keys;
  D8  experiment name is explicit or a deterministic digest of the model
      architecture (never Python object hash).

Checkpoint/resume: the global model is persisted per round
with the round-start payload schema {'state_dict', 'update_name', 'n_epoch'}
— the wire format IS the checkpoint layout, as in the reference.
"""

from __future__ import annotations

import asyncio
import hashlib
import logging
import os
import time
from typing import Dict, Optional

import torch
from aiohttp import web

from baton_amd.control.registry import ClientRegistry
from baton_amd.control.rounds import RoundInProgress, RoundState
from baton_amd.control.wire import decode_payload, encode_payload
from baton_amd.fed.aggregate import fedavg_, weighted_loss_history
from baton_amd.utils.config import BatonConfig
from baton_amd.utils.json_clean import json_clean

log = logging.getLogger("baton.manager")


def model_digest(model: torch.nn.Module) -> str:
    """Deterministic architecture digest: sha1 over (key, dtype, shape) of
    the state dict. Replaces the reference's reliance on a user-defined
    ``__hash__`` (defect D8, demo.py:26-27 / manager.py:16)."""
    h = hashlib.sha1()
    for k, v in model.state_dict().items():
        h.update(f"{k}:{v.dtype}:{tuple(v.shape)};".encode())
    return h.hexdigest()[:12]


class Experiment:
    def __init__(
        self,
        model: torch.nn.Module,
        name: str,
        app: Optional[web.Application] = None,
        config: Optional[BatonConfig] = None,
    ):
        self.model = model
        self.name = name
        self.config = config or BatonConfig()
        cc = self.config.control
        self.registry = ClientRegistry(
            name, app, client_ttl=cc.client_ttl, cull_interval=cc.effective_cull_interval
        )
        self.rounds = RoundState(name)
        self._deadline_handle: Optional[asyncio.TimerHandle] = None
        self._round_started_at: Optional[float] = None
        self.metrics: Dict[str, float] = {
            "rounds_completed": 0,
            "total_samples": 0,
            "last_round_sec": 0.0,
        }
        if app is not None:
            self.register_handlers(app)

    # -- HTTP surface --------------------------------------------------------

    def register_handlers(self, app: web.Application) -> None:
        exp = self.name
        app.router.add_post(f"/{exp}/update", self.handle_update)
        app.router.add_get(f"/{exp}/start_round", self.handle_start_round)
        app.router.add_get(f"/{exp}/end_round", self.handle_end_round)
        app.router.add_get(f"/{exp}/loss_history", self.handle_loss_history)
        app.router.add_get(f"/{exp}/round_state", self.handle_round_state)
        app.router.add_get(f"/{exp}/metrics", self.handle_metrics)

    async def handle_start_round(self, request: web.Request) -> web.Response:
        try:
            n_epoch = int(request.query.get("n_epoch", self.config.train.n_epoch))
        except ValueError:
            raise web.HTTPBadRequest(text="bad n_epoch")
        try:
            started = await self.start_round(n_epoch)
        except RoundInProgress:
            # 423 Locked — parity with manager.py:61-63 (aiohttp has no
            # HTTPLocked class; return the status code directly)
            return web.Response(status=423, text="round already in progress")
        return web.json_response(
            {"started": started, **json_clean(self.rounds.state_dict_meta())}
        )

    async def handle_end_round(self, request: web.Request) -> web.Response:
        if not self.rounds.in_progress:
            return web.json_response(
                {"ended": False, **json_clean(self.rounds.state_dict_meta())}
            )
        await self.end_round(reason="manual")
        return web.json_response(
            {"ended": True, **json_clean(self.rounds.state_dict_meta())}
        )

    async def handle_update(self, request: web.Request) -> web.Response:
        record = self.registry.verify_request(request)
        if record is None:
            raise web.HTTPUnauthorized(text="unknown client or bad key")
        body = await request.read()
        try:
            meta, tensors = decode_payload(body)
        except (ValueError, KeyError) as e:
            raise web.HTTPBadRequest(text=f"bad payload: {e}")
        update_name = meta.get("update_name", "")
        if not self.rounds.is_current(update_name):
            # 410 Gone — stale update (parity with manager.py:101-103)
            raise web.HTTPGone(text=f"stale update {update_name}")
        self.rounds.record(
            record.client_id,
            {
                "state_dict": tensors,
                "n_samples": int(meta.get("n_samples", 0)),
                "loss_history": list(meta.get("loss_history", [])),
                # True when the RCCL data plane already averaged the model
                # (parallel/gpu_worker.py) — end_round then copies instead
                # of re-averaging
                "aggregated": bool(meta.get("aggregated", False)),
            },
        )
        record.last_update = time.monotonic()
        record.num_updates += 1
        log.info(
            "update from %s (%d samples; %d left)",
            record.client_id,
            int(meta.get("n_samples", 0)),
            self.rounds.clients_left,
        )
        if self.rounds.clients_left == 0:
            await self.end_round(reason="complete")
        return web.json_response({"ok": True})

    async def handle_loss_history(self, request: web.Request) -> web.Response:
        # Defect D1 fixed: the reference returned a nonexistent attribute
        # (manager.py:48-49); this returns the real per-epoch series.
        return web.json_response({"loss_history": self.rounds.loss_history})

    async def handle_round_state(self, request: web.Request) -> web.Response:
        return web.json_response(json_clean(self.rounds.state_dict_meta()))

    async def handle_metrics(self, request: web.Request) -> web.Response:
        return web.json_response(
            {
                **self.metrics,
                "round_log": self.rounds.round_log[-20:],
                "n_clients": len(self.registry),
            }
        )

    # -- round lifecycle -----------------------------------------------------

    def start(self) -> None:
        self.registry.start()

    async def stop(self) -> None:
        self._cancel_deadline()
        await self.registry.stop()

    async def start_round(self, n_epoch: Optional[int] = None) -> bool:
        """Begin a federated round: snapshot membership, broadcast the global
        model, arm the deadline. Returns False if there were no clients
        (parity with manager.py:74-76)."""
        n_epoch = n_epoch if n_epoch is not None else self.config.train.n_epoch
        if len(self.registry) == 0:
            log.warning("start_round: no clients registered")
            return False
        update_name = self.rounds.begin(set(self.registry.client_ids))
        self._round_started_at = time.monotonic()
        payload = encode_payload(
            {"update_name": update_name, "n_epoch": n_epoch},
            self.model.state_dict(),
        )
        results = await self.registry.notify_clients("round_start", payload)
        accepted = 0
        for cid, status in results.items():
            if status == 200:
                self.rounds.client_started(cid)
                accepted += 1
            else:
                self.rounds.client_failed(cid)
        log.info("round %s: %d/%d clients accepted", update_name, accepted, len(results))
        if accepted == 0:
            self.rounds.finish(reason="nobody-started")
            return False
        self._arm_deadline()
        return True

    def _arm_deadline(self) -> None:
        deadline = self.config.control.round_deadline
        if deadline is None:
            return
        loop = asyncio.get_event_loop()
        self._deadline_handle = loop.call_later(
            deadline, lambda: asyncio.ensure_future(self._deadline_fired())
        )

    def _cancel_deadline(self) -> None:
        if self._deadline_handle is not None:
            self._deadline_handle.cancel()
            self._deadline_handle = None

    async def _deadline_fired(self) -> None:
        self._deadline_handle = None
        if self.rounds.in_progress:
            log.warning(
                "round %s deadline fired with %d clients missing",
                self.rounds.update_name,
                self.rounds.clients_left,
            )
            await self.end_round(reason="deadline")

    async def end_round(self, reason: str = "complete") -> None:
        """Aggregate whatever responded into the global model.

        Policy (defect D7 made explicit): with partial_policy='partial',
        responders are sample-weight-averaged (matching the reference's
        manual end_round semantics, manager.py:118-126); with 'abort', a
        round that is not complete discards its responses."""
        self._cancel_deadline()
        incomplete = self.rounds.clients_left > 0
        responses = self.rounds.finish(reason=reason)
        if not responses:
            log.warning("end_round(%s): no responses — model unchanged", reason)
            return
        if incomplete and self.config.control.partial_policy == "abort":
            log.warning("end_round(%s): incomplete round aborted by policy", reason)
            return
        weights = [max(r["n_samples"], 0) for r in responses.values()]
        if sum(weights) <= 0:
            weights = [1.0] * len(responses)
        if self.config.control.aggregation_mode == "rccl":
            # RCCL data-plane mode (SERVER-side switch, never client
            # metadata): the clients already hold the weighted mean
            # (reduce+broadcast over xGMI); rank 0 shipped it — copy.
            carrier = next(
                (r for r in responses.values() if len(r["state_dict"])), None
            )
            if carrier is not None:
                self.model.load_state_dict(carrier["state_dict"])
                self.rounds.loss_history.extend(carrier["loss_history"])
            else:
                log.warning("rccl round carried no state_dict — model unchanged")
        else:
            if any(r.get("aggregated") for r in responses.values()):
                # a client claimed its state_dict is pre-aggregated, but this
                # experiment is in fedavg mode: ignore the claim (it would
                # let one client overwrite the global model) and weight-
                # average every reported state_dict as usual.
                log.warning(
                    "client-asserted 'aggregated' flag ignored: experiment "
                    "aggregation_mode is 'fedavg'"
                )
            sds = [r["state_dict"] for r in responses.values()]
            fedavg_(self.model.state_dict(), sds, weights)
            self.rounds.loss_history.extend(
                weighted_loss_history(
                    [r["loss_history"] for r in responses.values()], weights
                )
            )
        elapsed = time.monotonic() - (self._round_started_at or time.monotonic())
        self.metrics["rounds_completed"] += 1
        self.metrics["total_samples"] += sum(weights)
        self.metrics["last_round_sec"] = elapsed
        log.info(
            "round aggregated: %d clients, %d samples, %.3fs",
            len(responses),
            int(sum(weights)),
            elapsed,
        )
        self._save_checkpoint()

    # -- checkpoint / resume ---------------------------------------------------

    def _checkpoint_path(self) -> Optional[str]:
        d = self.config.checkpoint_dir
        if not d:
            return None
        os.makedirs(d, exist_ok=True)
        return os.path.join(d, f"{self.name}.ckpt")

    def _save_checkpoint(self) -> None:
        path = self._checkpoint_path()
        if path is None:
            return
        payload = encode_payload(
            {
                "update_name": f"update_{self.name}_{self.rounds.round_index:05d}",
                "n_epoch": self.config.train.n_epoch,
                "round_index": self.rounds.round_index,
                "loss_history": self.rounds.loss_history,
            },
            self.model.state_dict(),
        )
        tmp = path + ".tmp"
        with open(tmp, "wb") as f:
            f.write(payload)
        os.replace(tmp, path)

    def load_checkpoint(self) -> bool:
        """Resume the global model + round counter from disk. Returns True
        if a checkpoint was loaded."""
        path = self._checkpoint_path()
        if path is None or not os.path.exists(path):
            return False
        with open(path, "rb") as f:
            meta, tensors = decode_payload(f.read())
        self.model.load_state_dict(tensors)
        self.rounds.round_index = int(meta.get("round_index", 0))
        self.rounds.loss_history = list(meta.get("loss_history", []))
        log.info("resumed %s at round %d", self.name, self.rounds.round_index)
        return True


class Manager:
    """Experiment factory bound to an aiohttp app (parity with
    manager.py:10-18)."""

    def __init__(self, app: Optional[web.Application] = None, config: Optional[BatonConfig] = None):
        self.app = app
        self.config = config or BatonConfig()
        self.experiments: Dict[str, Experiment] = {}

    def register_experiment(
        self, model: torch.nn.Module, name: Optional[str] = None
    ) -> Experiment:
        name = name or getattr(model, "name", None) or model_digest(model)
        if name in self.experiments:
            raise ValueError(f"experiment {name!r} already registered")
        exp = Experiment(model, name, app=self.app, config=self.config)
        self.experiments[name] = exp
        exp.start()
        return exp

    async def stop(self) -> None:
        for exp in self.experiments.values():
            await exp.stop()
