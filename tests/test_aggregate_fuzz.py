"""Property-based FedAvg math (hypothesis): sample-weighted aggregation
vs a float64 numpy oracle over arbitrary shapes/dtypes/weights — the
exact computation of reference manager.py:119-126 with the D4 buffer
policy (0-dim + integer buffers copied from the heaviest client)."""

from collections import OrderedDict

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from baton_amd.fed.aggregate import fedavg_


@st.composite
def clients(draw):
    n_clients = draw(st.integers(min_value=1, max_value=5))
    n_tensors = draw(st.integers(min_value=1, max_value=4))
    shapes = [
        tuple(draw(st.lists(st.integers(0, 4), min_size=0, max_size=3)))
        for _ in range(n_tensors)
    ]
    dtypes = [
        draw(st.sampled_from([torch.float32, torch.bfloat16, torch.int64]))
        for _ in range(n_tensors)
    ]
    sds = []
    for _ in range(n_clients):
        sd = OrderedDict()
        for i, (shp, dt) in enumerate(zip(shapes, dtypes)):
            if dt.is_floating_point:
                sd[f"p{i}"] = torch.randn(shp).to(dt)
            else:
                sd[f"p{i}"] = torch.randint(0, 50, shp, dtype=dt)
        sds.append(sd)
    weights = [draw(st.integers(min_value=1, max_value=1000))
               for _ in range(n_clients)]
    return sds, weights


@settings(max_examples=60, deadline=None)
@given(cw=clients())
def test_fedavg_matches_float64_oracle(cw):
    sds, weights = cw
    target = OrderedDict((k, v.clone()) for k, v in sds[0].items())
    fedavg_(target, sds, weights)
    total = float(sum(weights))
    heaviest = max(range(len(weights)), key=lambda i: weights[i])
    for k in target:
        if target[k].is_floating_point():
            oracle = sum(
                sd[k].double() * (w / total) for sd, w in zip(sds, weights)
            )
            got = target[k].double()
            tol = 1e-6 if target[k].dtype == torch.float32 else 3e-2
            assert torch.allclose(got, oracle, rtol=tol, atol=tol), k
        else:
            # integer buffers: copied from the heaviest client, not averaged
            assert torch.equal(target[k], sds[heaviest][k]), k
