"""Property-based robustness tests (hypothesis): the wire codec must
round-trip arbitrary payloads exactly and reject arbitrary garbage with
ValueError only (the reference's pickle wire is RCE-by-design, SURVEY.md
§2.5 D6 — ours must be total)."""

from collections import OrderedDict

import pytest
import torch
from hypothesis import given, settings, strategies as st

from baton_amd.control.wire import decode_payload, encode_payload

json_scalars = st.one_of(
    st.none(),
    st.booleans(),
    st.integers(min_value=-(2**53), max_value=2**53),
    st.floats(allow_nan=False, allow_infinity=False, width=32),
    st.text(max_size=20),
)
metas = st.dictionaries(
    st.text(min_size=1, max_size=16),
    st.one_of(json_scalars, st.lists(json_scalars, max_size=4)),
    max_size=6,
)

DTYPES = [torch.float32, torch.bfloat16, torch.float16, torch.int64, torch.int32]


@st.composite
def state_dicts(draw):
    n = draw(st.integers(min_value=0, max_value=4))
    out = OrderedDict()
    for i in range(n):
        name = f"t{i}." + draw(st.text(min_size=1, max_size=8).filter(str.strip))
        shape = draw(st.lists(st.integers(0, 5), min_size=0, max_size=3))
        dtype = draw(st.sampled_from(DTYPES))
        if dtype.is_floating_point:
            t = torch.randn(shape).to(dtype)
        else:
            t = torch.randint(-100, 100, shape, dtype=dtype)
        out[name] = t
    return out


@settings(max_examples=60, deadline=None)
@given(meta=metas, sd=state_dicts())
def test_roundtrip_exact(meta, sd):
    blob = encode_payload(meta, sd)
    meta2, sd2 = decode_payload(blob)
    assert meta2 == meta
    assert list(sd2) == list(sd)
    for k in sd:
        assert sd2[k].dtype == sd[k].dtype
        assert sd2[k].shape == sd[k].shape
        assert torch.equal(sd2[k], sd[k])


@settings(max_examples=120, deadline=None)
@given(data=st.binary(max_size=256))
def test_garbage_never_escapes_valueerror(data):
    try:
        decode_payload(data)
    except ValueError:
        pass  # the only allowed failure mode


@settings(max_examples=60, deadline=None)
@given(cut=st.integers(min_value=0, max_value=200), sd=state_dicts())
def test_truncation_never_escapes_valueerror(cut, sd):
    blob = encode_payload({"update_name": "u", "n_samples": 3}, sd)
    cut = min(cut, len(blob))
    truncated = blob[:cut]
    try:
        meta, got = decode_payload(truncated)
    except ValueError:
        return
    # a prefix that still decodes must at least reproduce the meta
    assert meta.get("update_name") == "u" or cut < len(blob)
