"""Client registry + liveness: the manager side of the membership layer.

Re-designed equivalent of the reference's ClientManager
(/root/reference/client_manager.py:14-150): issues client_id/key on
registration, tracks heartbeats, TTL-culls stale clients on a periodic task,
fans out round-start notifications concurrently, drops clients on 404 /
connection-refused, and authenticates requests. HTTP routes and status-code
contract preserved:

  GET /{exp}/register   -> {'client_id', 'key'}            (200)
  GET /{exp}/heartbeat  -> 200 | 401 unknown client/key
  GET /{exp}/clients    -> JSON client table, secrets stripped

Differences from the reference (by design):
  * keys via ``secrets`` (defect D6), not ``random.sample``
  * ``stop()`` actually cancels the cull task
  * notify failures return per-client results so the round layer can make an
    explicit membership decision (defect D7)
"""

from __future__ import annotations

import logging
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

import aiohttp
from aiohttp import web

from baton_amd.utils import PeriodicTask, json_clean, new_client_id, new_key

log = logging.getLogger("baton.registry")


@dataclass
class ClientRecord:
    """Per-client record — field parity with client_manager.py:100-109."""

    client_id: str
    key: str
    remote: str
    port: int
    url: Optional[str] = None
    last_heartbeat: float = field(default_factory=time.monotonic)
    last_update: Optional[float] = None
    num_updates: int = 0

    @property
    def base_url(self) -> str:
        if self.url:
            return self.url.rstrip("/")
        return f"http://{self.remote}:{self.port}"

    def to_public_dict(self) -> dict:
        return json_clean(
            {
                "client_id": self.client_id,
                "remote": self.remote,
                "port": self.port,
                "url": self.url,
                "last_heartbeat": self.last_heartbeat,
                "last_update": self.last_update,
                "num_updates": self.num_updates,
            }
        )


class ClientRegistry:
    def __init__(
        self,
        experiment: str,
        app: Optional[web.Application] = None,
        client_ttl: float = 300.0,
        cull_interval: Optional[float] = None,
        clock: Callable[[], float] = time.monotonic,
    ):
        self.experiment = experiment
        self.client_ttl = client_ttl
        self.clock = clock
        self._clients: Dict[str, ClientRecord] = {}
        self._session: Optional[aiohttp.ClientSession] = None
        self._cull_task = PeriodicTask(
            self.cull_clients,
            cull_interval if cull_interval is not None else client_ttl / 2,
            name=f"cull[{experiment}]",
        )
        if app is not None:
            self.register_handlers(app)

    # -- lifecycle ---------------------------------------------------------

    def start(self) -> None:
        self._cull_task.start()

    async def stop(self) -> None:
        await self._cull_task.stop()
        if self._session is not None:
            await self._session.close()
            self._session = None

    async def session(self) -> aiohttp.ClientSession:
        if self._session is None or self._session.closed:
            self._session = aiohttp.ClientSession()
        return self._session

    # -- dict-style access (parity with client_manager.py:80-84) -----------

    def __len__(self) -> int:
        return len(self._clients)

    def __contains__(self, client_id: str) -> bool:
        return client_id in self._clients

    def __getitem__(self, client_id: str) -> ClientRecord:
        return self._clients[client_id]

    def __iter__(self):
        return iter(self._clients.values())

    @property
    def client_ids(self) -> List[str]:
        return list(self._clients)

    # -- HTTP handlers ------------------------------------------------------

    def register_handlers(self, app: web.Application) -> None:
        exp = self.experiment
        app.router.add_get(f"/{exp}/register", self.handle_register)
        app.router.add_get(f"/{exp}/heartbeat", self.handle_heartbeat)
        app.router.add_get(f"/{exp}/clients", self.handle_clients)

    async def handle_register(self, request: web.Request) -> web.Response:
        remote = request.remote or "127.0.0.1"
        try:
            port = int(request.query.get("port", 8080))
        except ValueError:
            raise web.HTTPBadRequest(text="bad port")
        url = request.query.get("url") or None
        record = self.register(remote=remote, port=port, url=url)
        return web.json_response({"client_id": record.client_id, "key": record.key})

    async def handle_heartbeat(self, request: web.Request) -> web.Response:
        client_id = request.query.get("client_id", "")
        key = request.query.get("key", "")
        if not self.heartbeat(client_id, key):
            raise web.HTTPUnauthorized(text="unknown client or bad key")
        return web.json_response({"ok": True})

    async def handle_clients(self, request: web.Request) -> web.Response:
        return web.json_response(
            {"clients": [c.to_public_dict() for c in self._clients.values()]}
        )

    # -- core operations ----------------------------------------------------

    def register(self, remote: str, port: int, url: Optional[str] = None) -> ClientRecord:
        record = ClientRecord(
            client_id=new_client_id(self.experiment),
            key=new_key(),
            remote=remote,
            port=port,
            url=url,
            last_heartbeat=self.clock(),
        )
        self._clients[record.client_id] = record
        log.info("registered %s from %s:%s", record.client_id, remote, port)
        return record

    def heartbeat(self, client_id: str, key: str) -> bool:
        record = self._clients.get(client_id)
        if record is None or record.key != key:
            return False
        record.last_heartbeat = self.clock()
        return True

    def verify_request(self, request: web.Request) -> Optional[ClientRecord]:
        """Authenticate a request by query params (API parity with
        client_manager.py:144-150; the query-string transport is kept for
        wire compatibility — deploy behind TLS)."""
        client_id = request.query.get("client_id", "")
        key = request.query.get("key", "")
        record = self._clients.get(client_id)
        if record is None or record.key != key:
            return None
        return record

    async def cull_clients(self) -> None:
        now = self.clock()
        stale = [
            cid
            for cid, rec in self._clients.items()
            if now - rec.last_heartbeat > self.client_ttl
        ]
        for cid in stale:
            log.info("culling stale client %s", cid)
            self._clients.pop(cid, None)

    def drop(self, client_id: str) -> None:
        if self._clients.pop(client_id, None) is not None:
            log.info("dropped client %s", client_id)

    # -- fan-out ------------------------------------------------------------

    async def notify_client(
        self,
        record: ClientRecord,
        endpoint: str,
        data: bytes,
        timeout: float = 30.0,
    ) -> Tuple[str, Optional[int]]:
        """POST ``data`` to one client. Returns (client_id, status) with
        status None on connection failure. Drops the client on 404 or
        connection-refused (parity with client_manager.py:58-61)."""
        sess = await self.session()
        url = f"{record.base_url}/{self.experiment}/{endpoint}"
        params = {"client_id": record.client_id, "key": record.key}
        try:
            async with sess.post(
                url,
                params=params,
                data=data,
                timeout=aiohttp.ClientTimeout(total=timeout),
            ) as resp:
                status = resp.status
        except (aiohttp.ClientError, OSError, TimeoutError):
            log.warning("notify %s: connection failed — dropping", record.client_id)
            self.drop(record.client_id)
            return record.client_id, None
        if status == 404:
            log.warning("notify %s: 404 — dropping", record.client_id)
            self.drop(record.client_id)
        return record.client_id, status

    async def notify_clients(
        self, endpoint: str, data: bytes, timeout: float = 30.0
    ) -> Dict[str, Optional[int]]:
        """Concurrent fan-out to every registered client
        (client_manager.py:35-47). Returns {client_id: status|None}."""
        import asyncio

        records = list(self._clients.values())
        results = await asyncio.gather(
            *(self.notify_client(r, endpoint, data, timeout) for r in records)
        )
        return dict(results)
