"""Stat accumulation + cluster-wide aggregation.

Capability parity with the reference's examples/common/__init__.py:
StatMean/StatSum, delta-based global stat allreduce with requeue-on-error
(GlobalStatsAccumulator), RunningMeanStd.
"""
import copy
import dataclasses
import logging

import torch


@dataclasses.dataclass
class StatMean:
    value: float = 0
    n: int = 0

    def result(self):
        if self.n == 0:
            return None
        return self.value / self.n

    def __sub__(self, other):
        return StatMean(self.value - other.value, self.n - other.n)

    def __iadd__(self, other):
        if isinstance(other, StatMean):
            self.value += other.value
            self.n += other.n
        else:
            self.value += other
            self.n += 1
        return self

    def reset(self):
        self.value = 0
        self.n = 0

    def __repr__(self):
        return repr(self.result())


@dataclasses.dataclass
class StatSum:
    value: float = 0

    def result(self):
        return self.value

    def __sub__(self, other):
        return StatSum(self.value - other.value)

    def __iadd__(self, other):
        self.value += other.value if isinstance(other, StatSum) else other
        return self

    def reset(self):
        pass

    def __repr__(self):
        return repr(self.result())


class GlobalStatsAccumulator:
    """Reduce per-peer stat deltas across the group, asynchronously.

    Deltas since the previous reduce are queued and summed cluster-wide with
    group.all_reduce; on error the sent delta is re-queued so no counts are
    lost (same protocol as the reference)."""

    def __init__(self, rpc_group, global_stats):
        self.rpc_group = rpc_group
        self.global_stats = global_stats
        self.reduce_future = None
        self.queued = None
        self.sent = None
        self.prev_stats = None

    @staticmethod
    def add_stats(dst, src):
        for k, v in dst.items():
            v += src[k]
        return dst

    def _enqueue(self, stats):
        if self.queued is None:
            self.queued = copy.deepcopy(stats)
        else:
            self.add_stats(self.queued, stats)

    def reduce(self, stats):
        if self.reduce_future is not None and self.reduce_future.done():
            exc = self.reduce_future.exception()
            if exc is not None:
                logging.info("global stats accumulation error: %s", exc)
                self._enqueue(self.sent)
            else:
                self.add_stats(self.global_stats, self.reduce_future.result())
            self.reduce_future = None

        diff = stats
        if self.prev_stats is not None:
            diff = {k: v - self.prev_stats[k] for k, v in stats.items()}
        self._enqueue(diff)
        self.prev_stats = copy.deepcopy(stats)

        if self.reduce_future is None:
            self.sent = self.queued
            self.queued = None
            self.reduce_future = self.rpc_group.all_reduce(
                "global stats", copy.deepcopy(self.sent), self.add_stats
            )

    def reset(self):
        if self.prev_stats is not None:
            for v in self.prev_stats.values():
                v.reset()


class RunningMeanStd:
    """Parallel-algorithm running mean/variance (Chan et al.)."""

    def __init__(self, epsilon=1e-4, shape=()):
        self.mean = torch.zeros(shape, dtype=torch.float64)
        self.var = torch.ones(shape, dtype=torch.float64)
        self.count = epsilon

    def update(self, x):
        bmean = torch.mean(x, axis=0)
        bvar = torch.var(x, axis=0)
        bcount = x.shape[0]
        delta = bmean - self.mean
        tot = self.count + bcount
        self.mean = self.mean + delta * bcount / tot
        m2 = self.var * self.count + bvar * bcount + torch.square(delta) * self.count * bcount / tot
        self.var = m2 / tot
        self.count = tot
