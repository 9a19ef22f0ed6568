"""Local training loop — the per-client hot loop of a federated round.

The reference's equivalent is ``Model.train`` (/root/reference/demo.py:29-49):
per epoch, randperm minibatching, zero_grad -> forward -> loss -> backward ->
SGD step, per-epoch mean loss appended to a history. This re-design keeps
those semantics and adds:

  * optimizer choice (fused SGD / fused Adam from baton_amd.ops on GPU);
  * FedProx proximal term mu/2 * ||theta - theta_global||^2 (BASELINE.json
    config 5) applied as grad += mu * (theta - theta_global) at step time;
  * optional hipGraph capture of the steady-state step (runtime/graph.py);
  * device placement + bf16 compute support.
"""

from __future__ import annotations

import logging
from typing import Callable, List, Optional, Sequence

import torch

from baton_amd.utils.config import TrainConfig
from baton_amd.utils.progress import EpochProgress
from baton_amd.utils.tracing import trace_scope

log = logging.getLogger("baton.trainer")


class LocalTrainer:
    """Callable trainer: ``trainer(model, data, n_epoch) -> loss_history``.

    ``data`` is the tuple from ``ExperimentWorker.get_data()`` — positional
    tensors whose first dim is the sample dim; the last tensor is the target.
    ``loss_fn`` maps (model_output, target) -> scalar loss.
    """

    def __init__(
        self,
        config: Optional[TrainConfig] = None,
        loss_fn: Optional[Callable] = None,
        device: Optional[torch.device] = None,
        seed: Optional[int] = None,
    ):
        self.config = config or TrainConfig()
        self.loss_fn = loss_fn or torch.nn.functional.mse_loss
        self.device = device
        self.seed = seed
        self._graph_step = None   # set when hipGraph capture is active

    def make_optimizer(self, model: torch.nn.Module):
        from baton_amd.ops.optim import make_optimizer

        return make_optimizer(model.parameters(), self.config)

    def __call__(
        self, model: torch.nn.Module, data: Sequence[torch.Tensor], n_epoch: int
    ) -> List[float]:
        cfg = self.config
        device = self.device or next(model.parameters()).device
        model = model.to(device)
        *inputs, target = [t.to(device) for t in data]
        n = target.shape[0]
        opt = self.make_optimizer(model)

        # FedProx: snapshot the round's global weights once
        global_params = None
        if cfg.fedprox_mu > 0:
            global_params = [p.detach().clone() for p in model.parameters()]

        gen = None
        if self.seed is not None:
            gen = torch.Generator().manual_seed(self.seed)

        was_training = model.training
        model.train()

        # hipGraph capture of the steady-state minibatch step (fixed batch
        # shape; tail batches run eager). FedProx modifies grads between
        # backward and step, which capture does not cover.
        graph_step = None
        if (cfg.use_hip_graph and device.type == "cuda" and len(inputs) == 1
                and cfg.fedprox_mu == 0 and n >= cfg.batch_size):
            from baton_amd.runtime.graph import GraphedTrainStep

            graph_step = GraphedTrainStep(
                model, opt, self.loss_fn,
                inputs[0][: cfg.batch_size], target[: cfg.batch_size],
            )

        loss_history: List[float] = []
        for epoch in range(n_epoch):
            perm = torch.randperm(n, generator=gen)
            # live progress display (reference parity: the epoch loop is
            # tqdm-wrapped with a running-loss postfix, utils.py:70-90 /
            # demo.py:36-39) — with the D5 bias fixed by RunningMean
            progress = EpochProgress(
                epoch, list(torch.split(perm, cfg.batch_size)),
                use_tqdm=cfg.progress,
            )
            for batch_idx in progress:
                if graph_step is not None and len(batch_idx) == cfg.batch_size:
                    idx = batch_idx.to(device)
                    loss = graph_step(inputs[0][idx], target[idx])
                    progress.update_loss(loss.item(), weight=len(batch_idx))
                    continue
                bx = [t[batch_idx] for t in inputs]
                by = target[batch_idx]
                opt.zero_grad(set_to_none=True)
                with trace_scope("fwd"):
                    out = model(*bx)
                    loss = self.loss_fn(out, by)
                with trace_scope("bwd"):
                    loss.backward()
                if global_params is not None:
                    with torch.no_grad():
                        for p, g in zip(model.parameters(), global_params):
                            if p.grad is not None:
                                p.grad.add_(p.detach() - g, alpha=cfg.fedprox_mu)
                with trace_scope("opt"):
                    opt.step()
                progress.update_loss(loss.item(), weight=len(batch_idx))
            loss_history.append(progress.mean_loss or 0.0)
        model.train(was_training)
        return loss_history
