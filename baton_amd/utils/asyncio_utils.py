"""Async scaffolding: periodic tasks and coroutine single-flight.

Re-designed equivalents of the reference's L0 utilities
(/root/reference/utils.py:11-20 ``ensure_no_collision``, :42-67
``PeriodicTask``): same capabilities, plus cancellation that actually awaits
the task and an immediate-first-fire option.
"""

from __future__ import annotations

import asyncio
import functools
import logging
from typing import Awaitable, Callable, Optional

log = logging.getLogger("baton.utils")


class PeriodicTask:
    """Run ``await func()`` every ``interval`` seconds until stopped.

    Unlike the reference (utils.py:42-67) this survives exceptions in
    ``func`` (logged, loop continues) and ``stop()`` awaits cancellation so
    no orphan tasks outlive an experiment.
    """

    def __init__(
        self,
        func: Callable[[], Awaitable[None]],
        interval: float,
        *,
        fire_immediately: bool = False,
        name: Optional[str] = None,
    ):
        self.func = func
        self.interval = interval
        self.fire_immediately = fire_immediately
        self.name = name or getattr(func, "__name__", "periodic")
        self._task: Optional[asyncio.Task] = None

    @property
    def running(self) -> bool:
        return self._task is not None and not self._task.done()

    def start(self) -> "PeriodicTask":
        if not self.running:
            self._task = asyncio.ensure_future(self._run())
        return self

    async def _run(self) -> None:
        if self.fire_immediately:
            await self._fire()
        while True:
            await asyncio.sleep(self.interval)
            await self._fire()

    async def _fire(self) -> None:
        try:
            await self.func()
        except asyncio.CancelledError:
            raise
        except Exception:  # noqa: BLE001 — keep the heartbeat alive
            log.exception("periodic task %s raised", self.name)

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except asyncio.CancelledError:
                pass
            self._task = None


def single_flight(coro_fn):
    """Decorator: concurrent calls collapse into one in-flight execution.

    Equivalent capability to the reference's ``ensure_no_collision``
    (utils.py:11-20): a call made while a previous call is still running is
    skipped (returns None) instead of piling up — used for registration and
    heartbeat coroutines.
    """
    lock = asyncio.Lock()

    @functools.wraps(coro_fn)
    async def wrapper(*args, **kwargs):
        if lock.locked():
            return None
        async with lock:
            return await coro_fn(*args, **kwargs)

    return wrapper
