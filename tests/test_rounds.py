"""RoundState machine unit tests (423/410 paths per SURVEY.md §2.4/§3.3)."""

import pytest

from baton_amd.control.rounds import (
    RoundInProgress,
    RoundNotInProgress,
    RoundState,
)


def test_round_naming():
    rs = RoundState("exp")
    name = rs.begin({"a", "b"})
    assert name == "update_exp_00000"
    rs.finish()
    assert rs.begin({"a"}) == "update_exp_00001"


def test_double_begin_raises():
    rs = RoundState("exp")
    rs.begin({"a"})
    with pytest.raises(RoundInProgress):
        rs.begin({"a"})


def test_finish_without_begin_raises():
    rs = RoundState("exp")
    with pytest.raises(RoundNotInProgress):
        rs.finish()
    with pytest.raises(RoundNotInProgress):
        rs.record("a", {})


def test_membership_and_completion():
    rs = RoundState("exp")
    rs.begin({"a", "b", "c"})
    assert rs.clients_left == 3
    rs.record("a", {"n_samples": 1})
    assert rs.clients_left == 2
    rs.client_failed("c")  # c never accepted round_start
    assert rs.clients_left == 1
    rs.record("b", {"n_samples": 2})
    assert rs.clients_left == 0
    responses = rs.finish()
    assert set(responses) == {"a", "b"}
    assert not rs.in_progress


def test_staleness_check():
    rs = RoundState("exp")
    name = rs.begin({"a"})
    assert rs.is_current(name)
    assert not rs.is_current("update_exp_99999")
    rs.finish()
    assert not rs.is_current(name)  # finished round is stale -> HTTP 410


def test_round_log_reason():
    rs = RoundState("exp")
    rs.begin({"a", "b"})
    rs.record("a", {})
    rs.finish(reason="deadline")
    assert rs.round_log[-1]["reason"] == "deadline"
    assert rs.round_log[-1]["responded"] == 1
    assert rs.round_log[-1]["members"] == 2


def test_state_dict_meta_is_jsonable():
    import json

    rs = RoundState("exp")
    rs.begin({"b", "a"})
    meta = rs.state_dict_meta()
    json.dumps(meta)
    assert meta["members"] == ["a", "b"]
    assert meta["in_progress"] is True
