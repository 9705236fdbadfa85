"""Zero-copy same-node GPU tensor RPC via hipIpc (dmabuf) handles.

The reference refuses CUDA tensors on the wire outright (rpc.cc:661-667
`fatal`); moolib_amd stages them through the CPU by default, and this
module provides the zero-copy path BASELINE.json names: wrap a device
tensor in `share()` and pass it through any RPC call — the receiver
(same node, any process) materializes a tensor aliasing the sender's HBM
through a hipIpc memory handle, no bytes copied.

Built on torch's CUDA-IPC storage sharing (torch.multiprocessing
reductions == hipIpcGetMemHandle/hipIpcOpenMemHandle + a shared refcount
file on ROCm; requires HSA_ENABLE_IPC_MODE_LEGACY=0 with the dmabuf
driver).

    # producer (replay server on GPU 0)
    rpc.define("get", lambda: ipc.share(hbm_resident_batch))
    # consumer (learner on GPU 1, same node)
    batch = client.sync("server", "get")   # torch.Tensor aliasing GPU 0 HBM

Lifetime: the producer's storage stays alive until every consumer drops
its alias (torch's IPC refcounting). Do not send handles across nodes.
"""
import os
import threading
import time

import torch
from torch.multiprocessing import reductions

# Same-process short-circuit: a process cannot hipIpcOpenMemHandle its own
# handle, so handles deserialized by the producing process resolve through
# this registry instead. reduce_tensor() pre-increments a producer-side IPC
# ref counter that only a consumer's rebuild/release would decrement, so the
# same-process path must release it explicitly or every share pins its
# storage forever (torch reductions.py does the same on its storage-cache
# hit path). A TTL bounds the registry itself; TTL eviction deliberately
# does NOT release the ref counter: the producer cannot know whether a
# cross-process consumer already materialized the handle (that consumer's
# rebuild adopted the counter and decrements it on free — releasing here
# too would double-decrement and free HBM a consumer still aliases). The
# cost is that a share that is never deserialized anywhere pins its
# storage until the producer process exits; shares exist to be consumed,
# so that leak is bounded and preferable to a GPU use-after-free.
_local = {}  # key -> (tensor, deadline, func, args)
_lock = threading.Lock()  # shares/materializes run on concurrent RPC threads
_counter = [0]
_TTL = 120.0


def _release_producer_ref(func, args):
    if getattr(func, "__name__", "") != "rebuild_cuda_tensor" or len(args) < 13:
        return
    storage_cls, device = args[4], args[6]
    ref_counter_handle, ref_counter_offset = args[11], args[12]
    try:
        storage_cls._release_ipc_counter(
            ref_counter_handle, ref_counter_offset, device=device
        )
    except (AttributeError, RuntimeError):
        pass


def _evict_locked():
    # Drops registry entries (and our strong tensor reference) only; see the
    # module comment for why the IPC ref counter is never released here.
    now = time.monotonic()
    for k in [k for k, ent in _local.items() if ent[1] < now]:
        _local.pop(k)


def _materialize(pid, key, func, args):
    if pid == os.getpid():
        # Pop: the consumer takes ownership, so serving a stream of fresh
        # batches in-process doesn't pin them all for the TTL.
        with _lock:
            ent = _local.pop(key, None)
        if ent is not None:
            _release_producer_ref(func, args)
            return ent[0]
    return func(*args)


class SharedCudaTensor:
    """Pickles into a hipIpc handle; unpickles into the aliasing tensor."""

    def __init__(self, tensor):
        if not tensor.is_cuda:
            raise ValueError("share() is for device tensors; CPU tensors ship as bytes anyway")
        self._tensor = tensor

    def __reduce__(self):
        func, args = reductions.reduce_tensor(self._tensor)
        with _lock:
            _evict_locked()
            _counter[0] += 1
            key = _counter[0]
            _local[key] = (self._tensor, time.monotonic() + _TTL, func, args)
        return (_materialize, (os.getpid(), key, func, args))

    def tensor(self):
        return self._tensor


def share(tensor):
    """Wrap a CUDA tensor so RPC serialization sends a hipIpc handle
    instead of the bytes. The result of deserialization IS the tensor."""
    return SharedCudaTensor(tensor)
