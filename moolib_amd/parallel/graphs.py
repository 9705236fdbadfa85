"""hipGraph capture helpers (torch.cuda.CUDAGraph == hipGraph on ROCm).

The IMPALA hot loops are launch-bound on MI355X: the eager actor forward is
~70 kernels of ~5-15 us each with comparable gaps between them, and the
learner step is ~900 launches. Shapes are fixed ([1,B] actor, [T+1,B]
learner), so both are captured once and replayed — the CDNA4-idiomatic
answer to launch overhead (no tracing compiler, no Triton).
"""
import logging

import torch

from moolib_amd.utils import nest


def _copy_nest(dst, src):
    # One multi-tensor launch instead of one copy kernel per input:
    # _foreach_copy_ handles mixed dtypes/shapes as long as each pair
    # matches elementwise, which graph static inputs do by construction.
    ds, ss, = [], []
    for d, s in zip(nest.flatten(dst), nest.flatten(src)):
        if isinstance(d, torch.Tensor):
            if d.is_cuda and d.shape == s.shape:
                ds.append(d)
                ss.append(s)
            else:
                d.copy_(s, non_blocking=True)
    if ds:
        torch._foreach_copy_(ds, ss, non_blocking=True)


class GraphedCall:
    """Capture fn(static_inputs) -> outputs after `warmup` eager calls.

    call(inputs) runs eagerly until enough warmups have happened, then
    captures and afterwards replays with inputs copied into static buffers.
    Outputs are the static output nest — consumers must copy what they keep
    before the next call. Falls back to eager permanently if capture fails.
    """

    def __init__(self, fn, warmup=3, name="graph", generators=()):
        self.fn = fn
        self.warmup = warmup
        self.name = name
        self.generators = list(generators)  # non-default RNGs used inside fn
        self.calls = 0
        self.graph = None
        self.static_in = None
        self.static_out = None
        self.failed = False

    def __call__(self, inputs):
        if self.failed or self.calls < self.warmup:
            self.calls += 1
            return self.fn(inputs)
        if self.graph is None:
            try:
                self.static_in = nest.map(
                    lambda t: t.clone() if isinstance(t, torch.Tensor) else t, inputs
                )
                torch.cuda.synchronize()
                g = torch.cuda.CUDAGraph()
                for gen in self.generators:
                    # dedicated (non-default) generators must be registered
                    # before capture so replays advance their offsets
                    g.register_generator_state(gen.graphsafe_get_state())
                # thread_local: background threads (the accumulator's
                # scheduler staging gradient buckets, RPC callbacks) keep
                # doing legitimate GPU work while we capture; the default
                # global mode lets their ops interleave into the capture,
                # which intermittently produced corrupted graphs that fault
                # at replay (HSA_STATUS_ERROR_EXCEPTION ~1 in 8 runs).
                with torch.cuda.graph(g, capture_error_mode="thread_local"):
                    self.static_out = self.fn(self.static_in)
                self.graph = g
                logging.info("captured hipGraph '%s'", self.name)
            except Exception as e:  # noqa: BLE001
                logging.warning("hipGraph capture failed for '%s' (%s); staying eager", self.name, e)
                self.failed = True
                self.graph = None
                if torch.cuda.is_available():
                    torch.cuda.synchronize()  # leave no half-captured state behind
                return self.fn(inputs)
        _copy_nest(self.static_in, inputs)
        self.graph.replay()
        return self.static_out
