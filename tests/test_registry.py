"""ClientRegistry unit tests with a fake clock (TTL cull, auth)."""

import asyncio

import pytest

from baton_amd.control.registry import ClientRegistry


class FakeClock:
    def __init__(self):
        self.t = 1000.0

    def __call__(self):
        return self.t


@pytest.fixture
def registry():
    clock = FakeClock()
    reg = ClientRegistry("exp", app=None, client_ttl=300.0, clock=clock)
    return reg, clock


def test_register_issues_identity(registry):
    reg, _ = registry
    rec = reg.register(remote="10.0.0.1", port=8081)
    assert rec.client_id.startswith("client_exp_")
    assert len(rec.key) == 32
    assert len(reg) == 1
    assert rec.client_id in reg
    assert reg[rec.client_id] is rec


def test_keys_are_unique(registry):
    reg, _ = registry
    ids = {reg.register("h", 1).client_id for _ in range(50)}
    keys = {rec.key for rec in reg}
    assert len(ids) == 50
    assert len(keys) == 50


def test_heartbeat_updates_and_rejects(registry):
    reg, clock = registry
    rec = reg.register("h", 1)
    clock.t += 100
    assert reg.heartbeat(rec.client_id, rec.key)
    assert rec.last_heartbeat == clock.t
    assert not reg.heartbeat(rec.client_id, "wrong-key")
    assert not reg.heartbeat("ghost", rec.key)


def test_ttl_cull(registry):
    reg, clock = registry
    rec1 = reg.register("h", 1)
    rec2 = reg.register("h", 2)
    clock.t += 200
    reg.heartbeat(rec2.client_id, rec2.key)
    clock.t += 150  # rec1 is now 350s stale (> 300), rec2 only 150s
    asyncio.run(reg.cull_clients())
    assert rec1.client_id not in reg
    assert rec2.client_id in reg


def test_base_url_prefers_explicit_url(registry):
    reg, _ = registry
    rec = reg.register("10.1.2.3", 9090)
    assert rec.base_url == "http://10.1.2.3:9090"
    rec2 = reg.register("x", 1, url="http://worker.example:7000/")
    assert rec2.base_url == "http://worker.example:7000"


def test_public_dict_strips_key(registry):
    reg, _ = registry
    rec = reg.register("h", 1)
    pub = rec.to_public_dict()
    assert "key" not in pub
    assert pub["client_id"] == rec.client_id
