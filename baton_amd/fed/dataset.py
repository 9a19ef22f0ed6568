"""Federated dataset utilities.

The reference's only data story is the ``get_data`` hook returning a tuple
of tensors (worker.py:126-127). This module provides the production
equivalent: a tensor-backed dataset shared across a node, partitioned into
per-client shards (IID or Dirichlet non-IID), with each GPU-client holding
its shard resident in HBM (288 GB per GPU — keep it on device, never
re-read from host).
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import torch

from baton_amd.fed.split import dirichlet_partition, iid_partition


class FederatedTensorDataset:
    """Tensors (same first dim) + a partition into client shards.

    >>> ds = FederatedTensorDataset((x, y), n_clients=8, split="dirichlet",
    ...                             alpha=0.1, label_index=1)
    >>> xs, ys = ds.shard(rank)        # this client's tensors
    """

    def __init__(
        self,
        tensors: Sequence[torch.Tensor],
        n_clients: int,
        split: str = "iid",
        alpha: float = 0.1,
        label_index: Optional[int] = None,
        seed: int = 0,
    ):
        n = tensors[0].shape[0]
        for t in tensors:
            if t.shape[0] != n:
                raise ValueError("all tensors need the same first dim")
        self.tensors = tuple(tensors)
        self.n_clients = n_clients
        if split == "iid":
            self.partition: List[torch.Tensor] = iid_partition(n, n_clients, seed)
        elif split == "dirichlet":
            if label_index is None:
                raise ValueError("dirichlet split needs label_index")
            labels = tensors[label_index]
            self.partition = dirichlet_partition(labels, n_clients, alpha, seed)
        else:
            raise ValueError(f"unknown split {split!r}")

    def shard(self, client: int, device=None) -> Tuple[torch.Tensor, ...]:
        idx = self.partition[client]
        out = tuple(t[idx] for t in self.tensors)
        if device is not None:
            out = tuple(t.to(device) for t in out)
        return out

    def shard_size(self, client: int) -> int:
        return len(self.partition[client])

    def sizes(self) -> List[int]:
        return [len(p) for p in self.partition]
