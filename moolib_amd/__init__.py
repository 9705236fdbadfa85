"""moolib_amd — an MI355X-native distributed RL / RPC framework.

A brand-new implementation of the capabilities of facebookresearch/moolib
(reference: py/moolib/__init__.py), built for AMD Instinct MI355X: the
peer-to-peer RPC layer runs on an epoll reactor with zero-copy tensor
frames; gradient all-reduce runs on RCCL over xGMI when peers share a
torch.distributed world (see moolib_amd.parallel); device math for the
IMPALA path is hand-written HIP/CDNA4 kernels (moolib_amd.ops).
"""
import asyncio
import atexit
import threading
import time

__version__ = "0.1.0"

import torch

from . import _core

# The gfx950 kernel extension must load wherever a GPU is present: a silent
# fallback to eager torch on an MI355X would defeat the point of the
# framework (the ops modules only use eager paths off-GPU / in explicitly
# disabled modes).
try:
    from . import _kernels  # noqa: F401
except ImportError as _e:  # pragma: no cover
    if torch.cuda.is_available():
        raise ImportError(
            "moolib_amd._kernels (gfx950 HIP kernels) failed to load on a GPU "
            "machine: %s — rebuild with `python setup.py build_ext --inplace`"
            % _e
        ) from _e
    _kernels = None

from ._core import (
    Broker,
    Future,
    Queue,
    RpcDeferredReturn,
    RpcError,
    create_uid,
    set_log_level,
    set_logging,
    set_max_threads,
)
from .utils import nest

# Group / AllReduce --------------------------------------------------------

Group = _core.Group


class AllReduce(Future):
    """Handle for an in-flight collective (reference src/group.h:493-499).

    `group.all_reduce(name, value, op)` returns this Future: `result()`
    blocks for the reduced value, `done()/exception()/await` as usual; a
    membership change cancels in-flight reductions with an error (retry
    under the new sync_id). Subclass of Future so isinstance checks work
    both ways."""


# Future / Queue awaitability ---------------------------------------------


def _future_await(self):
    if not self.done():
        loop = asyncio.get_running_loop()
        af = loop.create_future()

        def _transfer():
            if not af.cancelled():
                af.set_result(None)

        def _cb():
            try:
                loop.call_soon_threadsafe(_transfer)
            except RuntimeError:
                pass  # loop already closed

        self._add_done_callback(_cb)
        yield from af.__await__()
    exc = self.exception()
    if exc is not None:
        raise exc
    return self.result()


Future.__await__ = _future_await
Future.__iter__ = _future_await


def _queue_await(self):
    return self._pop_future().__await__()


def _queue_get(self):
    """Blocking pop (convenience; not in the reference API)."""
    return self._pop_future().result()


Queue.__await__ = _queue_await
Queue.__iter__ = _queue_await
Queue.get = _queue_get

_core.Batcher.__await__ = _queue_await
_core.Batcher.__iter__ = _queue_await

# Fused batcher copies: with a GPU present, every Batcher.stack/cat call
# moves all its tensor leaves in ONE _kernels.batched_copy launch instead
# of one runtime copy per leaf (the kernel falls back to copy_ per pair
# for layouts/devices it cannot express, so semantics are unchanged).
if _kernels is not None and torch.cuda.is_available():
    _core._set_batcher_fused_copy(_kernels.batched_copy)


# Rpc with batched defines --------------------------------------------------


class _BatchCollector:
    """Collects deferred calls until batch_size is reached, then fires.

    Implements the reference's define(batch_size=...) semantics
    (src/moolib.cc:1007-1178 + batch_utils): tensor leaves are stacked along
    a new dim 0 and moved to `device`; non-tensor leaves pass through from
    the first call; returned tensors are split back per caller.
    """

    def __init__(
        self,
        batch_size,
        device,
        process,
        dynamic=False,
        max_latency=0.01,
        min_latency=0.0005,
    ):
        self.batch_size = batch_size
        self.device = device
        self.process = process  # fn(batched_args, batched_kwargs, respond_all)
        self.dynamic = dynamic
        self.max_latency = max_latency
        self.min_latency = min_latency
        self.lock = threading.Lock()
        self.pending = []
        self.timer = None
        # Latency model (reference src/moolib.cc dynamic batching): the
        # collection window tracks the measured batch service time — while
        # one batch computes, the next one fills, so waiting ~proc_ema adds
        # throughput without adding pipeline latency. arrival_ema estimates
        # whether a partial batch is worth holding for more arrivals.
        self.proc_ema = None  # EMA of process() wall time
        self.arrival_ema = None  # EMA of inter-arrival gap
        self._last_arrival = None

    def _window(self):
        # Queue-style handlers only enqueue (service time ~0, invisible to
        # us): keep the full window so batches still form. Direct handlers
        # adapt the window to their measured service time.
        if not self.dynamic or self.proc_ema is None or self.proc_ema < 1e-3:
            return self.max_latency
        return min(max(self.proc_ema, self.min_latency), self.max_latency)

    def add(self, deferred, args, kwargs):
        fire = None
        now = time.monotonic()
        with self.lock:
            if self._last_arrival is not None:
                gap = now - self._last_arrival
                self.arrival_ema = (
                    gap
                    if self.arrival_ema is None
                    else 0.8 * self.arrival_ema + 0.2 * gap
                )
            self._last_arrival = now
            self.pending.append((deferred, args, kwargs))
            n = len(self.pending)
            full = n >= self.batch_size
            # Early flush on the MARGINAL rule: if even the NEXT arrival is
            # expected to take longer than the window, waiting only adds
            # latency. (Comparing the time to FILL the batch would flush
            # singles forever whenever batch_size exceeds the number of
            # concurrent callers.)
            stale = (
                self.dynamic
                and self.arrival_ema is not None
                and self.arrival_ema > self._window()
            )
            if full or stale:
                fire = self.pending
                self.pending = []
                if self.timer is not None:
                    self.timer.cancel()
                    self.timer = None
            elif self.dynamic and self.timer is None:
                self.timer = threading.Timer(self._window(), self._flush)
                self.timer.daemon = True
                self.timer.start()
        if fire:
            self._fire(fire)

    def _flush(self):
        with self.lock:
            fire = self.pending
            self.pending = []
            self.timer = None
        if fire:
            self._fire(fire)

    def _fire(self, batch):
        n = len(batch)
        args0, kwargs0 = batch[0][1], batch[0][2]

        def stack_leaves(leaves):
            if all(isinstance(x, torch.Tensor) for x in leaves):
                t = torch.stack(leaves)
                if self.device is not None:
                    t = t.to(self.device)
                return t
            return leaves[0]

        batched_args = nest.map_many(stack_leaves, *[b[1] for b in batch]) if args0 else ()
        batched_kwargs = (
            nest.map_many(stack_leaves, *[b[2] for b in batch]) if kwargs0 else {}
        )

        def respond_all(result):
            for i, (deferred, _, _) in enumerate(batch):
                out_i = nest.map(
                    lambda t, i=i: t[i].cpu() if isinstance(t, torch.Tensor) else t, result
                )
                deferred(out_i)

        t0 = time.monotonic()
        try:
            self.process(tuple(batched_args), dict(batched_kwargs), respond_all, n)
        except Exception as e:  # noqa: BLE001
            # Propagate the real handler exception to every caller in the
            # batch; otherwise they all get the generic "deferred return
            # dropped without a response" when the deferreds are destroyed.
            msg = "%s: %s" % (type(e).__name__, e)
            for deferred, _, _ in batch:
                try:
                    deferred.error(msg)
                except Exception:  # noqa: BLE001 — already responded
                    pass
            return
        dt = time.monotonic() - t0
        with self.lock:
            self.proc_ema = dt if self.proc_ema is None else 0.8 * self.proc_ema + 0.2 * dt


class Rpc(_core.Rpc):
    """Named RPC peer (reference API: src/moolib.cc Rpc bindings)."""

    def define(self, name, func, batch_size=None, device=None, dynamic_batching=False):
        if batch_size is None:
            if device is None:
                self._define_raw(name, func)
            else:

                def moved(*args, **kwargs):
                    args = nest.map(
                        lambda t: t.to(device) if isinstance(t, torch.Tensor) else t, args
                    )
                    kwargs = nest.map(
                        lambda t: t.to(device) if isinstance(t, torch.Tensor) else t, kwargs
                    )
                    return func(*args, **kwargs)

                self._define_raw(name, moved)
            return

        def process(bargs, bkwargs, respond_all, n):
            respond_all(func(*bargs, **bkwargs))

        collector = _BatchCollector(batch_size, device, process, dynamic=dynamic_batching)
        self.define_deferred_raw(name, lambda d, *a, **kw: collector.add(d, a, kw))

    def define_deferred(self, name, func, batch_size=None, device=None, dynamic_batching=False):
        if batch_size is None:
            self.define_deferred_raw(name, func)
            return

        def process(bargs, bkwargs, respond_all, n):
            func(respond_all, *bargs, **bkwargs)

        collector = _BatchCollector(batch_size, device, process, dynamic=dynamic_batching)
        self.define_deferred_raw(name, lambda d, *a, **kw: collector.add(d, a, kw))

    def define_queue(self, name, batch_size=None, device=None, dynamic_batching=False):
        if batch_size is None:
            return self.define_queue_raw(name)
        queue = Queue()

        def process(bargs, bkwargs, respond_all, n):
            queue.enqueue((respond_all, bargs, bkwargs))

        collector = _BatchCollector(batch_size, device, process, dynamic=dynamic_batching)
        self.define_deferred_raw(name, lambda d, *a, **kw: collector.add(d, a, kw))
        return queue


# Training components (bound as the C++ layer grows) -----------------------

try:
    from ._core import Accumulator, Batcher  # noqa: F401
except ImportError:  # pragma: no cover — during staged bring-up
    pass

from ._core import EnvStepperFuture  # noqa: F401
from ._core import EnvPool as _EnvPoolCore


class _DispatchedStep:
    """Future for a step whose CUDA action is still in flight to the CPU.

    result() waits for the dispatcher to hand the action to the workers
    (i.e. the real _core future to exist), then delegates."""

    __slots__ = ("_event", "_holder")

    def __init__(self):
        self._event = threading.Event()
        self._holder = []

    def _resolve(self, fut_or_exc):
        self._holder.append(fut_or_exc)
        self._event.set()

    def result(self):
        self._event.wait()
        v = self._holder[0]
        if isinstance(v, BaseException):
            raise v
        return v.result()


class EnvPool(_EnvPoolCore):
    """EnvPool with asynchronous CUDA action staging.

    A CUDA action tensor (the actor forward's output, usually still being
    computed when step() is called) is copied to a persistent pinned slot
    non_blocking and handed to a dispatcher thread that waits on a HIP
    event before waking the workers — the act loop never blocks on the
    device (the reference stages through a pinned tensor but then does a
    synchronous stream sync inline, src/env.cc:309-319). CPU actions take
    the direct path unchanged.
    """

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._staging = {}  # batch_index -> [pinned, torch.cuda.Event]
        self._dispatch_q = None
        self._dispatcher = None

    def _ensure_dispatcher(self):
        if self._dispatcher is None:
            import queue as _queue

            self._dispatch_q = _queue.SimpleQueue()
            self._dispatcher = threading.Thread(
                target=self._dispatch_loop, daemon=True, name="envpool-dispatch"
            )
            self._dispatcher.start()

    def __del__(self):
        q = getattr(self, "_dispatch_q", None)
        if q is not None:
            q.put(None)  # unblock and end the dispatcher thread

    def _dispatch_loop(self):
        while True:
            item = self._dispatch_q.get()
            if item is None:
                return
            b, pinned, ev, lazy = item
            try:
                ev.synchronize()  # D2H copy into the pinned slot has landed
                lazy._resolve(_EnvPoolCore.step(self, b, pinned))
            except BaseException as e:  # noqa: BLE001 — surfaced via result()
                lazy._resolve(e)

    def step(self, batch_index, action):
        if not (isinstance(action, torch.Tensor) and action.is_cuda):
            return super().step(batch_index, action)
        self._ensure_dispatcher()
        slot = self._staging.get(batch_index)
        if slot is None:
            pinned = torch.empty(
                action.shape, dtype=torch.int64, device="cpu"
            ).pin_memory()
            slot = [pinned, torch.cuda.Event()]
            self._staging[batch_index] = slot
        pinned, ev = slot
        pinned.copy_(action.detach().to(torch.int64), non_blocking=True)
        ev.record()
        lazy = _DispatchedStep()
        self._dispatch_q.put((batch_index, pinned, ev, lazy))
        return lazy


class EnvStepper:
    """Client handle for stepping an EnvPool's batches.

    API parity with the reference's EnvStepper (src/env.h:456-490); in this
    implementation EnvPool itself is steppable, so this is a thin view.
    """

    def __init__(self, pool):
        self._pool = pool

    def step(self, batch_index, action):
        return self._pool.step(batch_index, action)


# Real env-server role (reference src/env.h:363-453): EnvRunner(create_env)
# .start(shm_name) attaches to an EnvPool created elsewhere with
# shm_name=... + external_workers=True and serves one worker slot.
from ._core import EnvRunner  # noqa: F401


atexit.register(_core._shutdown_all)

__all__ = [
    "Accumulator",
    "AllReduce",
    "Batcher",
    "Broker",
    "EnvPool",
    "EnvRunner",
    "EnvStepper",
    "EnvStepperFuture",
    "Future",
    "Group",
    "Queue",
    "Rpc",
    "RpcDeferredReturn",
    "RpcError",
    "create_uid",
    "set_log_level",
    "set_logging",
    "set_max_threads",
]
