"""Collective-backend integration for the Accumulator's gradient data plane.

MI355X design (see BASELINE.json north star): when every accumulator peer is
a rank of one torch.distributed job (one process per GPU, backend "nccl" ==
RCCL on ROCm), the gradient bucket is reduced IN PLACE on-device by
ncclAllReduce over xGMI — the flat bucket never visits the CPU. The RPC tree
(C++ side) remains the elastic / cross-node fallback.

The reference stages every gradient through pinned CPU memory and reduces
over its TCP allreduce tree (reference: src/accumulator.cc:847-1003); this
module is the re-designed data plane.
"""
import torch


def reduce_hook(process_group=None):
    """Build a local-reduce hook backed by torch.distributed.all_reduce.

    The hook is called by the C++ Accumulator with the flat on-device
    gradient bucket; it starts an async SUM allreduce and returns a poll
    function the accumulator calls from update() until the collective
    completes.
    """
    import torch.distributed as dist

    def hook(flat):
        work = dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=process_group, async_op=True)

        def poll():
            if work.is_completed():
                work.wait()  # records stream dependency for the caller
                return True
            return False

        return poll

    return hook


def install_collective_backend(accumulator, process_group=None):
    """Route the accumulator's gradient allreduce through RCCL/xGMI (or any
    initialized torch.distributed backend)."""
    accumulator.set_local_reduce_hook(reduce_hook(process_group))


def bucket_size_bytes_for_xgmi(world_size, link_gbps=153.0, links=7):
    """Advisory bucket sizing for ring collectives over point-to-point xGMI.

    MI355X exposes 7 xGMI links per GPU at ~153 GB/s each; a ring allreduce
    is per-link bound, so buckets should be large enough that per-step launch
    latency amortizes: ~8 MiB per active ring is a good floor. With a single
    flat bucket (our default) this is moot — the whole model reduces in one
    collective — but sharded/overlapped setups can use this.
    """
    return max(8 << 20, int((link_gbps * 1e9 / 153.0) * 0.0005))
