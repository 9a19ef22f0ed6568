"""Round state machine.

Re-designed equivalent of the reference's UpdateManager
(/root/reference/update_manager.py:17-68): an in-progress latch, per-round
membership snapshot, response collection, loss history. Differences by
design:

  * the latch is explicit state, not an ``asyncio.Lock`` held across
    awaits — ``begin()`` raises :class:`RoundInProgress` (HTTP 423 at the
    manager, manager.py:61-63) instead of blocking;
  * an optional deadline: the round resolves after ``deadline`` seconds even
    if stragglers never report (defects D3/D7 — the reference hangs forever);
  * partial participation is an explicit policy recorded per round.

Round naming preserved: ``update_{exp}_{n:05d}`` (update_manager.py:26).
"""

from __future__ import annotations

import time
from typing import Any, Dict, List, Optional, Set


class RoundError(Exception):
    """Base round-state error (parity: UpdateException, update_manager.py:5)."""


class RoundInProgress(RoundError):
    """begin() while a round is active -> HTTP 423."""


class RoundNotInProgress(RoundError):
    """finish()/record() with no round active."""


class RoundState:
    def __init__(self, experiment: str):
        self.experiment = experiment
        self.round_index = 0
        self.update_name: Optional[str] = None
        self.in_progress = False
        self.started_at: Optional[float] = None
        self.members: Set[str] = set()
        self.responses: Dict[str, Dict[str, Any]] = {}
        self.loss_history: List[float] = []        # per-epoch weighted global loss
        self.round_log: List[dict] = []            # per-round metadata

    # -- queries -------------------------------------------------------------

    @property
    def clients_left(self) -> int:
        return len(self.members - set(self.responses))

    def is_current(self, update_name: str) -> bool:
        return self.in_progress and update_name == self.update_name

    def state_dict_meta(self) -> dict:
        """Cleaned round-state summary (what the reference's broken
        trigger_end_round intended to return — defect D1)."""
        return {
            "experiment": self.experiment,
            "round_index": self.round_index,
            "update_name": self.update_name,
            "in_progress": self.in_progress,
            "members": sorted(self.members),
            "responded": sorted(self.responses),
            "clients_left": self.clients_left,
        }

    # -- transitions ----------------------------------------------------------

    def begin(self, members: Set[str]) -> str:
        if self.in_progress:
            raise RoundInProgress(f"round {self.update_name} still in progress")
        self.in_progress = True
        self.update_name = f"update_{self.experiment}_{self.round_index:05d}"
        self.started_at = time.monotonic()
        self.members = set(members)
        self.responses = {}
        return self.update_name

    def client_started(self, client_id: str) -> None:
        if not self.in_progress:
            raise RoundNotInProgress("no round in progress")
        self.members.add(client_id)

    def client_failed(self, client_id: str) -> None:
        """A client did not accept round_start — remove it from membership so
        the round can complete without it."""
        if self.in_progress:
            self.members.discard(client_id)

    def record(self, client_id: str, data: Dict[str, Any]) -> None:
        if not self.in_progress:
            raise RoundNotInProgress("no round in progress")
        self.responses[client_id] = data

    def finish(self, reason: str = "complete") -> Dict[str, Dict[str, Any]]:
        """Close the round and return collected responses. The caller
        aggregates; policy for missing members is the caller's, recorded
        here per round."""
        if not self.in_progress:
            raise RoundNotInProgress("no round in progress")
        responses = self.responses
        elapsed = time.monotonic() - (self.started_at or time.monotonic())
        self.round_log.append(
            {
                "update_name": self.update_name,
                "members": len(self.members),
                "responded": len(responses),
                "reason": reason,
                "elapsed_sec": elapsed,
            }
        )
        self.in_progress = False
        self.round_index += 1
        self.members = set()
        self.responses = {}
        return responses
