"""L0 utility tests (RunningMean fixes defect D5; json_clean; config)."""

import asyncio
import datetime

import pytest

from baton_amd.utils import RunningMean, json_clean, new_client_id, new_key
from baton_amd.utils.asyncio_utils import PeriodicTask, single_flight
from baton_amd.utils.config import BatonConfig


def test_running_mean_unbiased():
    """The reference's update_loss yields 2.8 for [1,2,3,4] (defect D5);
    the true mean is 2.5."""
    rm = RunningMean()
    for x in [1.0, 2.0, 3.0, 4.0]:
        rm.update(x)
    assert rm.mean == pytest.approx(2.5)
    assert rm.count == 4


def test_running_mean_weighted():
    rm = RunningMean()
    rm.update(1.0, weight=3)
    rm.update(5.0, weight=1)
    assert rm.mean == pytest.approx(2.0)


def test_json_clean_strips_secrets_and_tensors():
    import torch

    d = {
        "key": "secret",
        "state_dict": {"w": torch.zeros(3)},
        "client_id": "c1",
        "when": datetime.datetime(2026, 1, 1),
        "nested": {"key": "secret2", "ok": [1, 2, (3, 4)]},
        "obj": object(),
    }
    out = json_clean(d)
    assert "key" not in out
    assert "state_dict" not in out
    assert out["client_id"] == "c1"
    assert out["when"].startswith("2026-01-01")
    assert "key" not in out["nested"]
    assert out["nested"]["ok"] == [1, 2, [3, 4]]
    assert isinstance(out["obj"], str)
    import json

    json.dumps(out)


def test_keys():
    assert len(new_key()) == 32
    assert new_key() != new_key()
    cid = new_client_id("exp")
    assert cid.startswith("client_exp_")
    assert len(cid.split("_")[-1]) == 6


def test_periodic_task_runs_and_stops():
    async def scenario():
        count = 0

        async def tick():
            nonlocal count
            count += 1

        pt = PeriodicTask(tick, 0.02)
        pt.start()
        await asyncio.sleep(0.15)
        await pt.stop()
        observed = count
        assert observed >= 3
        await asyncio.sleep(0.06)
        assert count == observed  # no ticks after stop

    asyncio.run(scenario())


def test_periodic_task_survives_exceptions():
    async def scenario():
        count = 0

        async def tick():
            nonlocal count
            count += 1
            raise RuntimeError("boom")

        pt = PeriodicTask(tick, 0.02)
        pt.start()
        await asyncio.sleep(0.1)
        await pt.stop()
        assert count >= 2  # kept firing after the exception

    asyncio.run(scenario())


def test_single_flight_collapses_concurrent_calls():
    async def scenario():
        running = 0
        max_running = 0

        @single_flight
        async def work():
            nonlocal running, max_running
            running += 1
            max_running = max(max_running, running)
            await asyncio.sleep(0.05)
            running -= 1
            return "done"

        results = await asyncio.gather(*(work() for _ in range(5)))
        assert max_running == 1
        assert results.count("done") == 1  # others skipped -> None

    asyncio.run(scenario())


def test_config_defaults_match_reference():
    cfg = BatonConfig()
    assert cfg.control.client_ttl == 300.0        # manager.py:22
    assert cfg.control.heartbeat_interval == 60.0  # worker.py:14
    assert cfg.control.port == 8080                # worker.py:14
    assert cfg.control.effective_cull_interval == 150.0  # client_manager.py:23
    assert cfg.train.n_epoch == 32                 # manager.py:55
    assert cfg.train.lr == 1e-3                    # demo.py:29
    assert cfg.train.batch_size == 32              # demo.py:29


def test_config_from_dict_and_toml(tmp_path):
    cfg = BatonConfig.from_dict(
        {"control": {"port": 9000}, "train": {"lr": 0.1, "optimizer": "adam"},
         "device": "cpu"}
    )
    assert cfg.control.port == 9000
    assert cfg.train.lr == 0.1
    assert cfg.device == "cpu"
    with pytest.raises(KeyError):
        BatonConfig.from_dict({"train": {"nonsense": 1}})

    toml = tmp_path / "cfg.toml"
    toml.write_text(
        "[control]\nport = 9001\n[train]\nn_epoch = 4\n[data_plane]\nbackend = 'gloo'\n"
    )
    cfg2 = BatonConfig.from_toml(str(toml))
    assert cfg2.control.port == 9001
    assert cfg2.train.n_epoch == 4
    assert cfg2.data_plane.backend == "gloo"
