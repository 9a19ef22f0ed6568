"""hipGraph capture of the per-round worker step.

North-star requirement (BASELINE.json): "the per-round worker step is
hipGraph-captured". ``torch.cuda.CUDAGraph`` on ROCm IS hipGraph capture
(hipStreamBeginCapture / hipGraphLaunch under the hood). Capturing the
steady-state minibatch step (forward + loss + backward + fused optimizer)
collapses its ~100 kernel launches into one ``hipGraphLaunch``.

Usage:
    step = GraphedTrainStep(model, opt, loss_fn, example_x, example_y)
    for batch:
        loss_dev = step(bx, by)       # device tensor, no host sync
    total = step.loss_sum()           # one sync at epoch end

Constraints inherited from graph capture: fixed batch shape (the trainer
pads/drops the tail batch), static input buffers (batch data is copied in
before replay), and the optimizer state must already exist (a warmup step
runs on a side stream before capture — also required so RCCL/caching
allocator state is initialized).
"""

from __future__ import annotations

from typing import Callable

import torch


class GraphedTrainStep:
    def __init__(
        self,
        model: torch.nn.Module,
        opt,
        loss_fn: Callable,
        example_x: torch.Tensor,
        example_y: torch.Tensor,
        warmup_steps: int = 2,
    ):
        assert example_x.is_cuda, "hipGraph capture needs GPU tensors"
        self.model = model
        self.opt = opt
        self.loss_fn = loss_fn
        self.static_x = example_x.clone()
        self.static_y = example_y.clone()
        self._loss_accum = torch.zeros((), device=example_x.device, dtype=torch.float32)

        # warmup on a side stream (allocator + cuDNN-analog state)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(warmup_steps):
                self.opt.zero_grad()
                loss = self.loss_fn(self.model(self.static_x), self.static_y)
                loss.backward()
                self.opt.step()
        torch.cuda.current_stream().wait_stream(s)

        self.graph = torch.cuda.CUDAGraph()
        self.opt.zero_grad()
        with torch.cuda.graph(self.graph):
            loss = self.loss_fn(self.model(self.static_x), self.static_y)
            loss.backward()
            self.opt.step()
            self._static_loss = loss.detach()
            self._loss_accum += self._static_loss.float()
            # zero grads inside the graph so the next replay starts clean
            self.opt.zero_grad(set_to_none=False)

    @property
    def batch_size(self) -> int:
        return self.static_x.shape[0]

    def __call__(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        """Replay the captured step on a new batch. Returns the (device)
        loss tensor of this step; no host synchronization."""
        self.static_x.copy_(x, non_blocking=True)
        self.static_y.copy_(y, non_blocking=True)
        self.graph.replay()
        return self._static_loss

    def reset_loss_sum(self) -> None:
        self._loss_accum.zero_()

    def loss_sum(self) -> float:
        """Host-sync read of the accumulated loss over replays."""
        return float(self._loss_accum.item())


class GraphedRound:
    """hipGraph capture of an ENTIRE local round (every minibatch of every
    local epoch) as one graph.

    The single-minibatch capture (GraphedTrainStep) measured slower than
    eager at production batch sizes: its per-replay copy-in + replay floor
    exceeds the launch latency it saves (r1 bench note). Capturing the
    whole round amortizes ONE replay over every launch in the round, and —
    because the round iterates fixed slices of HBM-resident data — needs
    no per-replay copies at all.

    ``round_fn`` must run the full eager round reading only resident
    tensors and return the last (device) loss tensor; collectives stay
    outside the capture (the caller aggregates after replay).
    """

    def __init__(self, round_fn: Callable, warmup_rounds: int = 2):
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(warmup_rounds):
                round_fn()
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self._static_loss = round_fn()

    def __call__(self) -> torch.Tensor:
        self.graph.replay()
        return self._static_loss
