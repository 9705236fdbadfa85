"""Memfd-backed CPU tensors with zero-copy same-machine RPC identity.

The reference keeps EVERY RPC buffer in memfd segments so an (fd, offset)
pair identifies any buffer across processes (src/memory/memfd.cc); its
shipped fast path still memcpy's through unix-socket iovecs, though. Our
equivalent is explicit and allocation-scoped: tensors created with
`memfd_tensor()` live in their own memfd segment, and when one is sent
over RPC to a peer on this machine the wire carries only the segment
identity — the receiver re-opens the producer's memfd through
/proc/<pid>/fd/<fd> (same-user capability; no SCM_RIGHTS plumbing needed)
and maps the SAME pages. Writes are visible both ways (MAP_SHARED), like
the reference's shared buffers.

    buf = moolib_amd.shm.memfd_tensor((1024, 1024))   # ordinary CPU tensor
    rpc.define("get", lambda: buf)     # same-node peer receives an alias

Non-memfd CPU tensors are unaffected (they ship as zero-copy iovec bytes,
which is also what the reference actually does on its fast path).

Lifetime: the producer's fd stays open while the tensor's storage lives
(finalizer-closed); the receiver's mapping keeps the pages alive on its
own after that. A receiver that materializes after the producer storage
died gets a clean error (stale /proc path).
"""
import ctypes
import os
import weakref

import torch

_libc = ctypes.CDLL(None, use_errno=True)
_MFD_CLOEXEC = 0x0001

# storage data_ptr -> (fd, nbytes); data_ptr of a live memfd mapping is
# stable, and entries are removed by the storage finalizer that also
# closes the fd.
_registry = {}


def _memfd_create(name):
    fd = _libc.memfd_create(name.encode(), _MFD_CLOEXEC)
    if fd < 0:
        raise OSError(ctypes.get_errno(), "memfd_create failed")
    return fd


def memfd_tensor(shape, dtype=torch.float32):
    """A CPU tensor whose storage is a dedicated memfd segment (MAP_SHARED)."""
    shape = tuple(int(s) for s in shape)
    numel = 1
    for s in shape:
        numel *= s
    nbytes = numel * torch.empty(0, dtype=dtype).element_size()
    fd = _memfd_create("moolib-amd-%d" % os.getpid())
    try:
        os.ftruncate(fd, max(nbytes, 1))
        flat = torch.from_file(
            "/proc/self/fd/%d" % fd, shared=True, size=numel, dtype=dtype
        )
    except BaseException:
        os.close(fd)
        raise
    t = flat.view(shape)
    key = flat.untyped_storage().data_ptr()
    _registry[key] = (fd, nbytes)

    def _drop(key=key, fd=fd):
        _registry.pop(key, None)
        try:
            os.close(fd)
        except OSError:
            pass

    weakref.finalize(flat.untyped_storage(), _drop)
    return t


def _identity(t):
    """(fd, nbytes) if t's storage is one of our memfd segments, else None."""
    ent = _registry.get(t.untyped_storage().data_ptr())
    return ent


def _materialize(pid, fd, nbytes, inode, dtype, shape, strides, offset_el):
    if pid == os.getpid():
        path = "/proc/self/fd/%d" % fd
    else:
        path = "/proc/%d/fd/%d" % (pid, fd)
    # Guard against fd-number recycling: if the producer dropped the
    # segment and the fd number now names something else, the inode (and
    # usually size) won't match — fail loudly instead of aliasing
    # unrelated memory.
    st = os.stat(path)
    if st.st_ino != inode or st.st_size < nbytes:
        raise RuntimeError(
            "memfd tensor expired: fd %d of pid %d no longer names the "
            "shared segment (inode %d != %d)" % (fd, pid, st.st_ino, inode)
        )
    esize = torch.empty(0, dtype=dtype).element_size()
    flat = torch.from_file(path, shared=True, size=nbytes // esize, dtype=dtype)
    return flat.as_strided(shape, strides, offset_el)


class SharedMemfdTensor:
    """Pickles into the segment identity; unpickles into an aliasing tensor."""

    def __init__(self, tensor, fd, nbytes):
        self._t = tensor
        self._fd = fd
        self._nbytes = nbytes

    def __reduce__(self):
        t = self._t
        return (
            _materialize,
            (
                os.getpid(),
                self._fd,
                self._nbytes,
                os.fstat(self._fd).st_ino,
                t.dtype,
                tuple(t.shape),
                tuple(t.stride()),
                t.storage_offset(),
            ),
        )

    def tensor(self):
        return self._t


def share_if_memfd(t):
    """Wrap a CPU tensor for identity-send when its storage is a memfd
    segment; None otherwise (caller falls back to byte transfer)."""
    if t.device.type != "cpu":
        return None
    ent = _registry.get(t.untyped_storage().data_ptr())
    if ent is None:
        return None
    return SharedMemfdTensor(t, ent[0], ent[1])
