"""RCCL/xGMI collective bandwidth benchmark (multi-rank, standalone).

The reference ships test/test_multinode_allreduce.cc — a chunked ring
allreduce over its RPC plane timed for sizes 400 -> ~2.6M floats. On MI355X
dense allreduce belongs to RCCL over xGMI (7 p2p links x ~153 GB/s per
GPU), so this benchmark times torch.distributed all_reduce /
reduce_scatter / all_gather across sizes and reports algorithm and bus
bandwidth — the numbers that size the Accumulator's gradient buckets.

Run (one rank per GPU):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/rccl_bench.py
CPU check (gloo): same command with --backend gloo on a CPU box.
"""
import argparse
import os
import time

import torch
import torch.distributed as dist


def bench_op(op, tensor, iters, device):
    def sync():
        if device.type == "cuda":
            torch.cuda.synchronize()

    for _ in range(3):
        op(tensor)
    sync()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        op(tensor)
    sync()
    dt = time.perf_counter() - t0
    dist.barrier()
    return dt / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--backend", default=None)
    ap.add_argument("--max-mb", type=float, default=256.0)
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    use_cuda = torch.cuda.is_available()
    backend = args.backend or ("nccl" if use_cuda else "gloo")
    if world < 2:
        raise SystemExit("run under torch.distributed.run with >=2 ranks")
    dist.init_process_group(backend, rank=rank, world_size=world)
    if use_cuda and backend == "nccl":
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        if local_rank >= torch.cuda.device_count():
            # Verified on MI355X/ROCm 7: RCCL (2.26) refuses duplicate-device
            # communicators ("Duplicate GPU detected"), so a shared-GPU RCCL
            # smoke is impossible by design — use --backend gloo for a
            # 1-GPU multi-rank plumbing check; real RCCL numbers need one
            # GPU per rank.
            raise SystemExit(
                "rank %d has no dedicated GPU (%d visible): RCCL requires one "
                "GPU per rank; rerun with --backend gloo for a plumbing smoke"
                % (local_rank, torch.cuda.device_count())
            )
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")
    dtype = torch.bfloat16 if args.dtype == "bf16" and backend == "nccl" else torch.float32
    esize = torch.tensor([], dtype=dtype).element_size()

    sizes = []
    n = 400
    max_elems = int(args.max_mb * 1e6 / esize)
    while n <= max_elems:
        sizes.append(n)
        n *= 8

    if rank == 0:
        print(
            f"# backend={backend} world={world} dtype={dtype} "
            f"(bus bw = algo bw * 2(w-1)/w for allreduce)"
        )
        print(f"{'elems':>12} {'MB':>9} {'allreduce':>12} {'red-scat':>12} {'all-gath':>12}  (GB/s bus)")
    for n in sizes:
        n8 = (n // world) * world or world
        x = torch.ones(n8, dtype=dtype, device=device)
        shard = torch.empty(n8 // world, dtype=dtype, device=device)
        t_ar = bench_op(lambda t: dist.all_reduce(t), x, args.iters, device)
        if backend == "nccl":
            t_rs = bench_op(
                lambda t: dist.reduce_scatter_tensor(shard, t), x, args.iters, device
            )
            t_ag = bench_op(
                lambda t: dist.all_gather_into_tensor(t, shard), x, args.iters, device
            )
        else:
            t_rs = t_ag = float("nan")
        nbytes = n8 * esize
        bus_ar = nbytes / t_ar * 2 * (world - 1) / world / 1e9
        bus_rs = nbytes / t_rs * (world - 1) / world / 1e9 if t_rs == t_rs else float("nan")
        bus_ag = nbytes / t_ag * (world - 1) / world / 1e9 if t_ag == t_ag else float("nan")
        if rank == 0:
            print(
                f"{n8:>12} {nbytes/1e6:>9.2f} {bus_ar:>12.1f} {bus_rs:>12.1f} {bus_ag:>12.1f}"
            )
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
