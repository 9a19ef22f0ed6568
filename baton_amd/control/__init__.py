from baton_amd.control.wire import encode_payload, decode_payload
from baton_amd.control.registry import ClientRegistry, ClientRecord
from baton_amd.control.rounds import (
    RoundState,
    RoundError,
    RoundInProgress,
    RoundNotInProgress,
)
from baton_amd.control.manager import Manager, Experiment
from baton_amd.control.worker import ExperimentWorker

__all__ = [
    "encode_payload",
    "decode_payload",
    "ClientRegistry",
    "ClientRecord",
    "RoundState",
    "RoundError",
    "RoundInProgress",
    "RoundNotInProgress",
    "Manager",
    "Experiment",
    "ExperimentWorker",
]
