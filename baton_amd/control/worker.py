"""Worker runtime: registration, heartbeat, round execution, reporting.

Re-designed equivalent of the reference's ExperimentWorker
(/root/reference/worker.py:12-127). Behavior preserved: self-registration
with the manager, periodic heartbeat with exponential backoff on manager
outage and automatic re-registration on 401 (worker.py:57-79), a
``POST /{exp}/round_start`` handler that loads the global weights and kicks
off local training, and ``report_update`` with the reference's update
payload schema (worker.py:111-117).

Fixed by design:
  D2  the busy latch is real — an overlapping round_start gets HTTP 409;
  D3  training runs on a worker thread (``run_in_executor``) so heartbeats
      keep flowing during long local epochs and the TTL cull never kills a
      client mid-round;
  D6  payloads are wire.py (no pickle).

User contract (parity with worker.py:126-127 / demo.py:29-59): subclass and
implement ``get_data() -> ((tensors...), n_samples)``; the model provides
``train_round(*data, n_epoch) -> loss_history`` (the reference calls it
``train``, which shadows ``nn.Module.train(mode)`` — both names accepted).
Alternatively pass a ``trainer`` callable (see runtime/local.py) which
receives (model, data, n_epoch) and returns a loss history.
"""

from __future__ import annotations

import asyncio
import logging
from typing import Any, Callable, List, Optional, Sequence, Tuple

import aiohttp
from aiohttp import web

import torch

from baton_amd.control.wire import decode_payload, encode_payload
from baton_amd.control.manager import model_digest
from baton_amd.utils import PeriodicTask, single_flight
from baton_amd.utils.config import BatonConfig

log = logging.getLogger("baton.worker")


class ExperimentWorker:
    def __init__(
        self,
        app: Optional[web.Application],
        model: torch.nn.Module,
        manager_url: str,
        port: int = 8080,
        url: Optional[str] = None,
        name: Optional[str] = None,
        trainer: Optional[Callable[..., List[float]]] = None,
        config: Optional[BatonConfig] = None,
        auto_register: bool = True,
    ):
        self.model = model
        self.manager_url = manager_url.rstrip("/")
        self.port = port
        self.url = url
        self.config = config or BatonConfig()
        self.experiment = name or getattr(model, "name", None) or model_digest(model)
        self.trainer = trainer

        self.client_id: Optional[str] = None
        self.key: Optional[str] = None
        self._busy = asyncio.Lock()           # real busy latch (defect D2)
        self._session: Optional[aiohttp.ClientSession] = None
        self._heartbeat_task = PeriodicTask(
            self._heartbeat_guarded,
            self.config.control.heartbeat_interval,
            name=f"heartbeat[{self.experiment}]",
        )
        self._backoff = 1.0
        self.rounds_run = 0
        self.last_loss: Optional[float] = None

        if app is not None:
            self.register_handlers(app)
            if auto_register:
                app.on_startup.append(self._on_startup)
                app.on_cleanup.append(self._on_cleanup)

    # -- lifecycle -----------------------------------------------------------

    async def _on_startup(self, app: web.Application) -> None:
        asyncio.ensure_future(self.register_with_manager())

    async def _on_cleanup(self, app: web.Application) -> None:
        await self.stop()

    async def stop(self) -> None:
        await self._heartbeat_task.stop()
        if self._session is not None:
            await self._session.close()
            self._session = None

    async def session(self) -> aiohttp.ClientSession:
        if self._session is None or self._session.closed:
            self._session = aiohttp.ClientSession()
        return self._session

    def register_handlers(self, app: web.Application) -> None:
        app.router.add_post(f"/{self.experiment}/round_start", self.handle_round_start)
        app.router.add_get(f"/{self.experiment}/status", self.handle_status)

    # -- registration / heartbeat ---------------------------------------------

    @property
    def registered(self) -> bool:
        return self.client_id is not None

    async def register_with_manager(self) -> bool:
        """GET /{exp}/register, store identity, start heartbeats. Retries
        with exponential backoff until the manager is reachable (parity with
        worker.py:40-55 + the backoff of worker.py:74-79)."""
        sess = await self.session()
        params = {"port": str(self.port)}
        if self.url:
            params["url"] = self.url
        while True:
            try:
                async with sess.get(
                    f"{self.manager_url}/{self.experiment}/register",
                    params=params,
                    timeout=aiohttp.ClientTimeout(total=10),
                ) as resp:
                    if resp.status == 200:
                        body = await resp.json()
                        self.client_id = body["client_id"]
                        self.key = body["key"]
                        self._backoff = 1.0
                        log.info("registered as %s", self.client_id)
                        self._heartbeat_task.start()
                        return True
                    log.warning("register got HTTP %d", resp.status)
            except (aiohttp.ClientError, OSError) as e:
                log.warning("register failed (%s); retrying in %.1fs", e, self._backoff)
            await asyncio.sleep(self._backoff)
            self._backoff = min(self._backoff * 2, 60.0)

    @single_flight
    async def _heartbeat_guarded(self) -> None:
        await self.heartbeat()

    async def heartbeat(self) -> None:
        if not self.registered:
            return
        sess = await self.session()
        try:
            async with sess.get(
                f"{self.manager_url}/{self.experiment}/heartbeat",
                params={"client_id": self.client_id, "key": self.key},
                timeout=aiohttp.ClientTimeout(total=10),
            ) as resp:
                if resp.status == 401:
                    # manager forgot us (TTL cull / restart) — re-register
                    log.warning("heartbeat 401 — re-registering")
                    self.client_id = self.key = None
                    asyncio.ensure_future(self.register_with_manager())
        except (aiohttp.ClientError, OSError) as e:
            log.warning("heartbeat failed: %s", e)

    # -- round execution -------------------------------------------------------

    async def handle_status(self, request: web.Request) -> web.Response:
        return web.json_response(
            {
                "client_id": self.client_id,
                "busy": self._busy.locked(),
                "rounds_run": self.rounds_run,
                "last_loss": self.last_loss,
            }
        )

    async def handle_round_start(self, request: web.Request) -> web.Response:
        # auth: the manager echoes the identity it issued (worker.py:93-96)
        qid = request.query.get("client_id", "")
        qkey = request.query.get("key", "")
        if qid != self.client_id or qkey != self.key:
            # 404 tells the manager to drop this client record; we also
            # re-register to get a fresh identity (worker.py:94-96)
            asyncio.ensure_future(self._reregister())
            raise web.HTTPNotFound(text="not my identity")
        if self._busy.locked():
            # 409 — real busy guard (defect D2: the reference's flag was
            # never set, worker.py:25/88)
            raise web.HTTPConflict(text="round already running")
        # Acquire HERE, before any await point: two round_start requests
        # interleaving at request.read() would otherwise both pass the
        # locked() check and run serially, the second on a stale global
        # model instead of getting the intended 409. With no suspension
        # between the check and this acquire (an uncontended asyncio.Lock
        # acquires synchronously), the latch is race-free on the
        # single-threaded event loop.
        await self._busy.acquire()
        try:
            body = await request.read()
            meta, state_dict = decode_payload(body)
            update_name = meta["update_name"]
            n_epoch = int(meta.get("n_epoch", 1))
        except (ValueError, KeyError) as e:
            self._busy.release()
            raise web.HTTPBadRequest(text=f"bad payload: {e}")
        except BaseException:
            self._busy.release()
            raise
        asyncio.ensure_future(self._run_round(state_dict, update_name, n_epoch))
        return web.json_response({"ok": True, "update_name": update_name})

    async def _reregister(self) -> None:
        self.client_id = self.key = None
        await self.register_with_manager()

    async def _run_round(self, state_dict, update_name: str, n_epoch: int) -> None:
        # The busy latch was acquired by handle_round_start (before its
        # first await — TOCTOU-free); this task owns and releases it.
        try:
            loop = asyncio.get_event_loop()
            # Training runs on an executor thread so the event loop —
            # heartbeats included — stays live (defect D3).
            n_samples, loss_history = await loop.run_in_executor(
                None, self._train_locally, state_dict, n_epoch
            )
            self.rounds_run += 1
            self.last_loss = loss_history[-1] if loss_history else None
            await self.report_update(update_name, n_samples, loss_history)
        except Exception:
            log.exception("round %s failed on this worker", update_name)
        finally:
            self._busy.release()

    def _train_locally(self, state_dict, n_epoch: int) -> Tuple[int, List[float]]:
        """Synchronous: load global weights, fetch data, run local epochs."""
        self.model.load_state_dict(state_dict)
        data, n_samples = self.get_data()
        if self.trainer is not None:
            loss_history = self.trainer(self.model, data, n_epoch)
        elif hasattr(self.model, "train_round"):
            loss_history = self.model.train_round(*data, n_epoch=n_epoch)
        elif type(self.model).train is not torch.nn.Module.train:
            # reference contract: model.train(*data, n_epoch) (demo.py:29)
            loss_history = self.model.train(*data, n_epoch=n_epoch)
        else:
            raise RuntimeError(
                "no trainer: pass trainer= or implement model.train_round()"
            )
        return n_samples, [float(x) for x in loss_history]

    async def report_update(
        self, update_name: str, n_samples: int, loss_history: List[float]
    ) -> None:
        """POST the update payload (schema parity with worker.py:111-117);
        on 401 re-register (worker.py:121-122)."""
        payload = encode_payload(
            {
                "update_name": update_name,
                "n_samples": int(n_samples),
                "loss_history": loss_history,
            },
            self.model.state_dict(),
        )
        sess = await self.session()
        try:
            async with sess.post(
                f"{self.manager_url}/{self.experiment}/update",
                params={"client_id": self.client_id, "key": self.key},
                data=payload,
                timeout=aiohttp.ClientTimeout(total=120),
            ) as resp:
                if resp.status == 401:
                    log.warning("report 401 — re-registering")
                    asyncio.ensure_future(self._reregister())
                elif resp.status == 410:
                    log.warning("report rejected: stale round %s", update_name)
                elif resp.status != 200:
                    log.warning("report got HTTP %d", resp.status)
        except (aiohttp.ClientError, OSError) as e:
            log.warning("report failed: %s", e)

    # -- user hook -------------------------------------------------------------

    def get_data(self) -> Tuple[Sequence[Any], int]:
        """Return ((tensors...), n_samples) — the user data hook
        (parity with worker.py:126-127)."""
        raise NotImplementedError
