"""Data partitioners for federated experiments.

Absent from the reference (each demo worker just generates random data,
demo.py:52-59). Needed for BASELINE.json config 5: non-IID Dirichlet(alpha)
label splits across clients.
"""

from __future__ import annotations

from typing import List

import torch


def iid_partition(n_samples: int, n_clients: int, seed: int = 0) -> List[torch.Tensor]:
    """Random equal split of indices [0, n_samples) into n_clients shards."""
    g = torch.Generator().manual_seed(seed)
    perm = torch.randperm(n_samples, generator=g)
    return [shard for shard in torch.chunk(perm, n_clients)]


def dirichlet_partition(
    labels: torch.Tensor,
    n_clients: int,
    alpha: float = 0.1,
    seed: int = 0,
    min_per_client: int = 1,
) -> List[torch.Tensor]:
    """Non-IID label-skewed split: for each class, sample proportions from
    Dirichlet(alpha) over clients and assign that class's indices
    accordingly (the standard LDA split of Hsu et al., used by the
    BASELINE.json ResNet-50 FedProx config).

    Returns a list of index tensors, one per client, each non-empty
    (re-sampled until every client has >= min_per_client samples, so a
    degenerate alpha cannot produce an empty federated client).
    """
    labels = labels.flatten().long()
    classes = labels.unique()
    g = torch.Generator().manual_seed(seed)
    for attempt in range(100):
        shards: List[List[torch.Tensor]] = [[] for _ in range(n_clients)]
        for c in classes.tolist():
            idx = (labels == c).nonzero(as_tuple=True)[0]
            idx = idx[torch.randperm(len(idx), generator=g)]
            # Dirichlet(alpha) via normalized Gamma draws
            props = torch._standard_gamma(
                torch.full((n_clients,), alpha), generator=g
            )
            props = props / props.sum().clamp_min(1e-12)
            counts = (props * len(idx)).floor().long()
            # distribute the remainder to the largest proportions
            rem = len(idx) - int(counts.sum())
            if rem > 0:
                order = torch.argsort(props, descending=True)
                for k in range(rem):
                    counts[order[k % n_clients]] += 1
            start = 0
            for ci in range(n_clients):
                n = int(counts[ci])
                if n > 0:
                    shards[ci].append(idx[start : start + n])
                start += n
        out = [
            torch.cat(s) if s else torch.empty(0, dtype=torch.long) for s in shards
        ]
        if all(len(o) >= min_per_client for o in out):
            return out
    raise RuntimeError(
        f"dirichlet_partition: could not give every client >= {min_per_client} "
        f"samples (alpha={alpha}, n_clients={n_clients})"
    )
