"""Running statistics for training loops.

The reference's ``EpochProgress.update_loss`` (/root/reference/utils.py:85-90)
computes a biased running mean (defect D5: weights don't sum to 1 — for
[1,2,3,4] it yields 2.8 instead of 2.5). ``RunningMean`` is the correct
Welford-style incremental mean.
"""

from __future__ import annotations

from typing import Iterable, Iterator, Optional


class RunningMean:
    """Numerically stable incremental mean: m += (x - m) / n."""

    __slots__ = ("mean", "count")

    def __init__(self) -> None:
        self.mean = 0.0
        self.count = 0

    def update(self, x: float, weight: int = 1) -> float:
        self.count += weight
        self.mean += (x - self.mean) * weight / self.count
        return self.mean

    def reset(self) -> None:
        self.mean = 0.0
        self.count = 0


class EpochProgress:
    """Batch iterator with a correct running-loss display.

    Capability parity with the reference's EpochProgress (utils.py:70-90,
    tqdm-wrapped); tqdm is optional here so headless runs stay clean.
    """

    def __init__(self, epoch: int, batches: Iterable, use_tqdm: bool = False):
        self.epoch = epoch
        self._mean = RunningMean()
        self._bar = None
        if use_tqdm:
            try:
                from tqdm import tqdm

                batches = tqdm(batches, desc=f"epoch {epoch}")
                self._bar = batches
            except ImportError:
                pass
        self._it: Iterator = iter(batches)

    def __iter__(self):
        return self

    def __next__(self):
        return next(self._it)

    def update_loss(self, loss: float, weight: int = 1) -> float:
        mean = self._mean.update(float(loss), weight=weight)
        if self._bar is not None:
            self._bar.set_postfix(loss=f"{mean:.5f}")
        return mean

    @property
    def mean_loss(self) -> Optional[float]:
        return self._mean.mean if self._mean.count else None
