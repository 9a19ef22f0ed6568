"""Server-side aggregation-mode gate.

A client-supplied ``aggregated: true`` flag must never flip the manager
into copy mode: in the default 'fedavg' aggregation_mode the manager
weight-averages every reported state_dict, so one hostile (but
authenticated) client cannot overwrite the global model wholesale.
"""

import asyncio
from collections import OrderedDict

import pytest
import torch

from baton_amd.control.manager import Experiment
from baton_amd.utils.config import BatonConfig


def _model():
    torch.manual_seed(0)
    return torch.nn.Linear(4, 1)


def _run_round(exp, responses):
    exp.rounds.begin(set(responses))
    for cid in responses:
        exp.rounds.client_started(cid)
    for cid, rec in responses.items():
        exp.rounds.record(cid, rec)
    asyncio.run(exp.end_round(reason="complete"))


def _sd(value):
    return OrderedDict(
        [("weight", torch.full((1, 4), value)), ("bias", torch.full((1,), value))]
    )


def test_client_asserted_aggregated_flag_is_ignored_in_fedavg_mode():
    exp = Experiment(_model(), "sec", app=None)
    assert exp.config.control.aggregation_mode == "fedavg"
    honest = {
        "state_dict": _sd(1.0), "n_samples": 100,
        "loss_history": [0.5], "aggregated": False,
    }
    # hostile client claims its weights are "already aggregated"
    hostile = {
        "state_dict": _sd(9.0), "n_samples": 100,
        "loss_history": [0.5], "aggregated": True,
    }
    _run_round(exp, {"honest": honest, "hostile": hostile})
    # equal sample counts -> plain mean (5.0), NOT the hostile copy (9.0)
    got = exp.model.state_dict()["weight"]
    assert torch.allclose(got, torch.full((1, 4), 5.0))


def test_rccl_mode_is_a_server_config_switch():
    cfg = BatonConfig()
    cfg.control.aggregation_mode = "rccl"
    exp = Experiment(_model(), "sec2", app=None, config=cfg)
    rank0 = {
        "state_dict": _sd(3.0), "n_samples": 100,
        "loss_history": [0.25], "aggregated": True,
    }
    rank1 = {
        "state_dict": OrderedDict(), "n_samples": 100,
        "loss_history": [0.25], "aggregated": True,
    }
    _run_round(exp, {"r0": rank0, "r1": rank1})
    got = exp.model.state_dict()["weight"]
    assert torch.allclose(got, torch.full((1, 4), 3.0))
    assert exp.rounds.loss_history == [0.25]
