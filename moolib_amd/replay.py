"""Distributed prioritized replay, resident in GPU HBM.

BASELINE.json config 5: "R2D2-style distributed prioritized replay resident
in 288 GB HBM (stresses tensor RPC)". A ReplayBuffer peer preallocates its
ring storage on the learner GPU — 288 GB HBM3E holds ~40M Atari frames
(uint8 84x84x4) without ever touching host memory — and serves add/sample
over the moolib RPC plane (tensors ride the wire out-of-band, staged
through pinned host memory on the GPU boundary).

Proportional prioritization (Schaul et al. 2016): P(i) ~ p_i^alpha, with
importance weights w_i = (N * P(i))^-beta / max w.
"""
import torch

from moolib_amd import ipc
from moolib_amd.utils import nest


class ReplayBuffer:
    """Ring buffer of fixed-shape sequence nests with proportional priorities.

    Can be used locally or served over RPC with `serve()`.
    """

    def __init__(self, capacity, device="cpu", alpha=0.6, beta=0.4):
        self.capacity = int(capacity)
        self.device = torch.device(device)
        self.alpha = alpha
        self.beta = beta
        self.storage = None  # nest of [capacity, ...] tensors
        self.priorities = torch.zeros(self.capacity, dtype=torch.float32, device=self.device)
        self.cursor = 0
        self.size = 0
        self.max_priority = 1.0

    def _ensure_storage(self, item):
        if self.storage is not None:
            return
        def alloc(t):
            return torch.empty(
                (self.capacity,) + tuple(t.shape), dtype=t.dtype, device=self.device
            )
        self.storage = nest.map(alloc, item)

    def add(self, item, priority=None):
        """Insert one sequence nest; returns its slot index."""
        item = nest.map(lambda t: torch.as_tensor(t), item)
        self._ensure_storage(item)
        idx = self.cursor
        for dst, src in zip(nest.flatten(self.storage), nest.flatten(item)):
            dst[idx].copy_(src.to(self.device, non_blocking=True))
        p = float(priority) if priority is not None else self.max_priority
        self.priorities[idx] = max(p, 1e-6) ** self.alpha
        self.max_priority = max(self.max_priority, p)
        self.cursor = (self.cursor + 1) % self.capacity
        self.size = min(self.size + 1, self.capacity)
        return idx

    def sample(self, batch_size):
        """Returns (batch nest [B, ...], indices [B], is_weights [B])."""
        if self.size == 0:
            raise RuntimeError("replay buffer is empty")
        probs = self.priorities[: self.size]
        idx = torch.multinomial(probs, batch_size, replacement=True)
        batch = nest.map(lambda t: t[idx], self.storage)
        psum = probs.sum()
        p = probs[idx] / psum
        w = (self.size * p).pow(-self.beta)
        w = w / w.max().clamp_min(1e-12)
        return batch, idx, w

    def update_priorities(self, indices, priorities):
        indices = torch.as_tensor(indices, device=self.device, dtype=torch.int64)
        priorities = torch.as_tensor(priorities, device=self.device, dtype=torch.float32)
        self.priorities[indices] = priorities.clamp_min(1e-6) ** self.alpha
        self.max_priority = max(self.max_priority, float(priorities.max()))

    def __len__(self):
        return self.size

    # ----------------------------------------------------------- serving

    def serve(self, rpc, name="replay"):
        """Expose this buffer on an Rpc peer."""

        def add(item, priority=None):
            return self.add(item, priority)

        def sample(batch_size):
            batch, idx, w = self.sample(batch_size)
            # Tensors cross the wire on the CPU; receivers move them where
            # they want. Same-node consumers use `.sample_ipc` instead.
            return (
                nest.map(lambda t: t.cpu(), batch),
                idx.cpu(),
                w.cpu(),
            )

        def sample_ipc(batch_size):
            # Zero-copy for same-node consumers: device tensors ship as
            # hipIpc handles (moolib_amd.ipc); idx/weights are tiny, CPU.
            batch, idx, w = self.sample(batch_size)
            if self.device.type == "cuda":
                batch = nest.map(ipc.share, batch)
            else:
                batch = nest.map(lambda t: t.cpu(), batch)
            return (batch, idx.cpu(), w.cpu())

        def update_priorities(indices, priorities):
            self.update_priorities(indices, priorities)

        def info():
            return {"size": self.size, "capacity": self.capacity}

        rpc.define(name + ".add", add)
        rpc.define(name + ".sample", sample)
        rpc.define(name + ".sample_ipc", sample_ipc)
        rpc.define(name + ".update_priorities", update_priorities)
        rpc.define(name + ".info", info)
        return self


class ReplayClient:
    """Client for a remote ReplayBuffer served on peer `server_name`."""

    def __init__(self, rpc, server_name, name="replay"):
        self.rpc = rpc
        self.server = server_name
        self.name = name

    def add(self, item, priority=None):
        return self.rpc.async_(self.server, self.name + ".add", item, priority)

    def sample(self, batch_size):
        return self.rpc.async_(self.server, self.name + ".sample", batch_size)

    def sample_ipc(self, batch_size):
        """Same-node zero-copy sample: device tensors arrive as hipIpc
        aliases of the server's HBM (do not use across nodes)."""
        return self.rpc.async_(self.server, self.name + ".sample_ipc", batch_size)

    def update_priorities(self, indices, priorities):
        return self.rpc.async_(
            self.server, self.name + ".update_priorities", indices, priorities
        )

    def info(self):
        return self.rpc.sync(self.server, self.name + ".info")
