"""LR/momentum schedules. OneCycle matches fastai fit_one_cycle semantics
(train.py:106-113: one_cycle with max_lr = 2*lr): cosine warmup from
max_lr/div over pct_start of steps, cosine anneal to max_lr/(div*1e4);
momentum (beta1) annealed 0.95 -> 0.85 -> 0.95."""
from __future__ import annotations

import math


class FlatSchedule:
    def __init__(self, lr: float):
        self.lr = lr

    def at(self, frac: float):
        return self.lr, None


class OneCycle:
    def __init__(self, max_lr: float, pct_start: float = 0.3, div: float = 25.0,
                 final_div: float = 25.0e4, moms=(0.95, 0.85)):
        self.max_lr, self.pct_start, self.div, self.final_div = max_lr, pct_start, div, final_div
        self.moms = moms

    @staticmethod
    def _cos(start: float, end: float, pct: float) -> float:
        return end + (start - end) / 2 * (math.cos(math.pi * pct) + 1)

    def at(self, frac: float):
        frac = min(max(frac, 0.0), 1.0)
        if frac < self.pct_start:
            p = frac / self.pct_start
            lr = self._cos(self.max_lr / self.div, self.max_lr, p)
            mom = self._cos(self.moms[0], self.moms[1], p)
        else:
            p = (frac - self.pct_start) / (1 - self.pct_start)
            lr = self._cos(self.max_lr, self.max_lr / self.final_div, p)
            mom = self._cos(self.moms[1], self.moms[0], p)
        return lr, mom
