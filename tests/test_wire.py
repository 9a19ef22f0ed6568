"""Wire-format round-trip tests (SURVEY.md §2.4 schemas, defect D6)."""

from collections import OrderedDict

import pytest
import torch

from baton_amd.control.wire import decode_payload, encode_payload


def _roundtrip(meta, sd):
    data = encode_payload(meta, sd)
    meta2, sd2 = decode_payload(data)
    return meta2, sd2


def test_roundtrip_round_start_schema():
    sd = OrderedDict(
        [("fc1.weight", torch.randn(1, 10)), ("fc1.bias", torch.randn(1))]
    )
    meta = {"update_name": "update_exp_00000", "n_epoch": 32}
    meta2, sd2 = _roundtrip(meta, sd)
    assert meta2 == meta
    assert list(sd2) == list(sd)
    for k in sd:
        assert torch.equal(sd[k], sd2[k])
        assert sd[k].dtype == sd2[k].dtype


def test_roundtrip_update_schema():
    sd = OrderedDict([("w", torch.randn(3, 3, dtype=torch.float64))])
    meta = {
        "update_name": "update_exp_00001",
        "n_samples": 320,
        "loss_history": [1.0, 0.5, 0.25],
    }
    meta2, sd2 = _roundtrip(meta, sd)
    assert meta2["n_samples"] == 320
    assert meta2["loss_history"] == [1.0, 0.5, 0.25]
    assert torch.equal(sd2["w"], sd["w"])


@pytest.mark.parametrize(
    "dtype",
    [torch.float32, torch.float16, torch.bfloat16, torch.int64, torch.int32,
     torch.uint8, torch.bool],
)
def test_roundtrip_dtypes(dtype):
    if dtype.is_floating_point:
        t = torch.randn(4, 5).to(dtype)
    elif dtype == torch.bool:
        t = torch.rand(4, 5) > 0.5
    else:
        t = torch.randint(0, 100, (4, 5), dtype=dtype)
    _, sd2 = _roundtrip({}, OrderedDict([("t", t)]))
    assert sd2["t"].dtype == dtype
    assert torch.equal(sd2["t"], t)


def test_roundtrip_zero_dim_and_empty():
    """Defect D4 shapes: BatchNorm num_batches_tracked is 0-dim int64."""
    sd = OrderedDict(
        [
            ("bn.num_batches_tracked", torch.tensor(7, dtype=torch.int64)),
            ("empty", torch.empty(0)),
        ]
    )
    _, sd2 = _roundtrip({}, sd)
    assert sd2["bn.num_batches_tracked"].item() == 7
    assert sd2["bn.num_batches_tracked"].shape == ()
    assert sd2["empty"].numel() == 0


def test_roundtrip_noncontiguous():
    t = torch.randn(8, 8).t()  # non-contiguous view
    _, sd2 = _roundtrip({}, OrderedDict([("t", t)]))
    assert torch.equal(sd2["t"], t)


def test_bad_magic_raises_not_executes():
    with pytest.raises(ValueError):
        decode_payload(b"PICKLED_EVIL_BYTES_" + b"x" * 64)


def test_truncated_raises():
    data = encode_payload({"a": 1}, OrderedDict([("t", torch.randn(10))]))
    with pytest.raises(Exception):
        decode_payload(data[: len(data) // 2])


def test_state_dict_loads_back_into_model():
    m = torch.nn.Linear(10, 1)
    _, sd2 = _roundtrip({}, OrderedDict(m.state_dict()))
    m2 = torch.nn.Linear(10, 1)
    m2.load_state_dict(sd2)
    for a, b in zip(m.parameters(), m2.parameters()):
        assert torch.equal(a, b)
