"""Stat accumulation + cluster-wide aggregation.

Capability parity with the reference's examples/common/__init__.py:
StatMean/StatSum, delta-based global stat allreduce with requeue-on-error
(GlobalStatsAccumulator), RunningMeanStd. The stat classes are our own
design around one protocol every stat must speak so it can travel through
the group allreduce: `+=` folds in a raw sample or a same-typed stat,
`-` produces the delta not yet shipped cluster-wide, `result()` renders,
`reset()` clears windowed state. Field names (`value`, `n`) are part of
the pickle wire format between peers — do not rename.
"""
import copy
import logging

import torch


class StatMean:
    """Mean over everything folded in since the last reset.

    Stored as (numerator, count) so instances add and subtract exactly —
    the delta protocol needs `a - b` then `c += delta` to be lossless.
    """

    __slots__ = ("value", "n")

    def __init__(self, value=0, n=0):
        self.value = value
        self.n = n

    def __iadd__(self, sample):
        inc = (sample.value, sample.n) if isinstance(sample, StatMean) else (sample, 1)
        self.value += inc[0]
        self.n += inc[1]
        return self

    def __sub__(self, baseline):
        return StatMean(self.value - baseline.value, self.n - baseline.n)

    def result(self):
        return None if self.n == 0 else self.value / self.n

    def reset(self):
        self.value, self.n = 0, 0

    def __repr__(self):
        return repr(self.result())

    def __eq__(self, other):
        if isinstance(other, StatMean):
            return self.value == other.value and self.n == other.n
        return NotImplemented


class StatSum:
    """Monotonic total; reset() is a no-op (lifetime counters survive
    stat-window resets, matching the reference's semantics)."""

    __slots__ = ("value",)

    def __init__(self, value=0):
        self.value = value

    def __iadd__(self, sample):
        self.value += sample.value if isinstance(sample, StatSum) else sample
        return self

    def __sub__(self, baseline):
        return StatSum(self.value - baseline.value)

    def result(self):
        return self.value

    def reset(self):
        pass

    def __repr__(self):
        return repr(self.result())

    def __eq__(self, other):
        if isinstance(other, StatSum):
            return self.value == other.value
        return NotImplemented


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
