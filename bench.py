#!/usr/bin/env python
"""Flagship benchmark: IMPALA Atari-ResNet env frames/sec (whole node).

Measures the metric BASELINE.json names — env frames/sec consumed for
training by the IMPALA/V-trace loop on the Atari deep-ResNet model at the
reference benchmark config (actor_batch_size 128 x 2 batches, 10 actor
processes, learn batch 32, unroll 20, virtual_batch_size 32, Adam 6e-4) on
synthetic 84x84x4 uint8 frames with random-init weights.

Single process per GPU; launched for N>1 by torch.distributed.run
(one rank per GPU over RCCL); the Accumulator's gradient bucket reduces via
dist.all_reduce (RCCL over xGMI) while membership/counts run on the moolib
RPC plane. Weak scaling: per-GPU actor+learner work is fixed as N grows.

One JSON line is printed by rank 0 at the end (driver contract).
"""
import argparse
import json
import os
import sys
import time


def run_r2d2(args):
    """BASELINE config 5: R2D2-style prioritized replay resident in HBM,
    sequences added and sampled over the tensor-RPC plane (hipIpc zero-copy
    when on GPU). Single-process; value = sequences sampled/s."""
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "examples"))
    import torch

    import r2d2_replay

    use_cuda = torch.cuda.is_available()
    device = args.device or ("cuda:0" if use_cuda else "cpu")
    seconds = max(8.0, min(float(args.steps), 60.0))
    m = r2d2_replay.run(device=device, seconds=seconds, ipc=use_cuda)
    line = {
        "metric": "replay sequences sampled/sec (R2D2 prioritized, HBM-resident)",
        "value": m["sampled_per_s"],
        "unit": "sequences/s",
        "n_gpus": 1 if use_cuda else 0,
        "steps": m["sampled"],
        "warmup": 0,
        "ms_per_step": 1000.0 / max(m["sampled_per_s"], 1e-9),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic",
        "config": {
            "model": "AtariNet-LSTM rollouts -> prioritized replay",
            "capacity_seqs": m["capacity"],
            "unroll": m["unroll"],
            "sample_batch": m["batch_size"],
            "num_envs": m["num_envs"],
            "ipc": m["ipc"],
            "frames_acted_per_s": m["frames_per_s"],
        },
    }
    import json

    print(json.dumps(line))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30, help="timed optimizer steps")
    ap.add_argument("--warmup", type=int, default=10, help="untimed warmup optimizer steps")
    ap.add_argument("--use-lstm", action="store_true")
    ap.add_argument("--device", default=None)
    ap.add_argument("--actor-batch-size", type=int, default=128)
    ap.add_argument("--num-actor-batches", type=int, default=2)
    ap.add_argument("--num-actor-cpus", type=int, default=10)
    ap.add_argument("--batch-size", type=int, default=32)
    ap.add_argument("--unroll-length", type=int, default=20)
    ap.add_argument("--virtual-batch-size", type=int, default=32)
    ap.add_argument("--max-seconds", type=float, default=1800.0)
    ap.add_argument("--backend", default=None, help="torch.distributed backend override")
    ap.add_argument("--breakdown", action="store_true", help="print per-phase wall time")
    ap.add_argument(
        "--config",
        default="impala",
        choices=["impala", "r2d2"],
        help="impala = the headline benchmark (default); r2d2 = BASELINE "
        "config 5, HBM-resident prioritized replay served over tensor RPC",
    )
    args = ap.parse_args()

    if args.config == "r2d2":
        return run_r2d2(args)

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    if world > 1:
        # 8-rank startup hardening: MIOpen's exhaustive conv find takes
        # ~15 s/process and all ranks contend on one user find-db file
        # lock; give each rank its own db and use the fast heuristic find.
        # (N=1 keeps the default exhaustive find — same conditions as the
        # single-GPU headline number.)
        os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")
        os.environ.setdefault(
            "MIOPEN_USER_DB_PATH", "/tmp/miopen-rank%d" % rank
        )

    # ---- 1. Fork env workers FIRST (clean pre-CUDA, pre-thread processes).
    import moolib_amd

    if os.environ.get("MOOLIB_AMD_LOG"):
        moolib_amd.set_log_level(os.environ["MOOLIB_AMD_LOG"])
    from moolib_amd.envs import SyntheticAtariEnv

    num_actions = 18
    envs = moolib_amd.EnvPool(
        lambda: SyntheticAtariEnv(num_actions=num_actions, mean_episode_len=1000),
        num_processes=args.num_actor_cpus,
        batch_size=args.actor_batch_size,
        num_batches=args.num_actor_batches,
    )

    # ---- 2. Now CUDA / torch.distributed.
    import torch

    use_cuda = torch.cuda.is_available()
    if args.device:
        device = args.device
    elif use_cuda:
        device = "cuda:%d" % local_rank
    else:
        device = "cpu"
    if use_cuda:
        dev_index = torch.device(device).index
        torch.cuda.set_device(dev_index if dev_index is not None else local_rank)

    dist = None
    if world > 1:
        import datetime

        import torch.distributed as dist

        backend = args.backend or ("nccl" if use_cuda else "gloo")
        dist.init_process_group(
            backend, rank=rank, world_size=world, timeout=datetime.timedelta(seconds=300)
        )

    # ---- 3. Control plane: rank 0 hosts the broker in-process.
    master_port = int(os.environ.get("MASTER_PORT", "0"))
    broker_port = master_port + 7 if master_port else 0
    broker = None
    if rank == 0:
        broker_rpc = moolib_amd.Rpc()
        broker_rpc.set_name("broker")
        broker = moolib_amd.Broker(broker_rpc)
        bound = broker_rpc.listen("127.0.0.1:%d" % broker_port)
        broker_addr = [a for a in bound if a.startswith("tcp://127")][0]
    if world > 1:
        obj = [broker_addr] if rank == 0 else [None]
        dist.broadcast_object_list(obj, src=0)
        broker_addr = obj[0]
    elif rank == 0:
        pass  # broker_addr already set

    from moolib_amd.impala import ImpalaConfig, ImpalaPeer

    cfg = ImpalaConfig(
        num_actions=num_actions,
        actor_batch_size=args.actor_batch_size,
        num_actor_batches=args.num_actor_batches,
        num_actor_cpus=args.num_actor_cpus,
        batch_size=args.batch_size,
        unroll_length=args.unroll_length,
        virtual_batch_size=args.virtual_batch_size,
        device=device,
        use_lstm=args.use_lstm,
        connect=broker_addr,
        local_name="rank%d" % rank,
        group_name="bench",
        lr_schedule=False,
        group_timeout=60.0,  # benchmark ranks never churn; avoid spurious eviction
    )
    peer = ImpalaPeer(
        cfg,
        envs=envs,
        use_collective_backend=world > 1,
        broker=broker,
    )

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    def run_steps(n, budget):
        t_end = time.time() + budget
        done = 0
        while done < n:
            if time.time() > t_end:
                print(
                    "STALL rank %d: %s | group %s active=%s | batcher=%d | events=%s"
                    % (
                        rank,
                        peer.accumulator.debug_state(),
                        peer.group.members(),
                        peer.group.active(),
                        peer.learn_batcher.size(),
                        {k: v.result() for k, v in peer.stats.items()},
                    ),
                    file=sys.stderr,
                    flush=True,
                )
                raise TimeoutError("benchmark stalled: %d/%d optimizer steps" % (done, n))
            ev = peer.step_once()
            if ev == "optimize":
                done += 1

    def smi_telemetry(tag):
        """Log GPU sclk/power/temp to stderr (bench-variance forensics:
        r1 saw a 47-63k f/s box-to-box spread for identical code)."""
        if not use_cuda or not os.environ.get("MOOLIB_AMD_BENCH_TELEMETRY"):
            return
        import subprocess

        try:
            out = subprocess.run(
                ["rocm-smi", "--showgpuclocks", "--showpower", "--showtemp"],
                capture_output=True, text=True, timeout=10,
            ).stdout
            keep = [
                l for l in out.splitlines()
                if any(k in l for k in ("sclk", "Power", "Temperature", "edge"))
            ]
            print("[telemetry %s rank %d]\n%s" % (tag, rank, "\n".join(keep)),
                  file=sys.stderr, flush=True)
        except Exception as e:  # noqa: BLE001 — telemetry must never kill the bench
            print("[telemetry %s failed: %s]" % (tag, e), file=sys.stderr)

    # ---- warmup ----
    smi_telemetry("pre-warmup")
    run_steps(args.warmup, args.max_seconds / 2)
    smi_telemetry("post-warmup")
    if args.breakdown:
        peer.profile = True
        peer.phase_times = {}

    # ---- timed region ----
    vbs_stat = peer.stats["virtual_batch_size"]
    v0 = vbs_stat.value  # running sum of global batch sizes over rounds
    barrier_sync()
    t0 = time.time()
    run_steps(args.steps, args.max_seconds)
    barrier_sync()
    elapsed = time.time() - t0
    smi_telemetry("post-timed")
    frames = (vbs_stat.value - v0) * args.unroll_length  # global frames consumed

    if dist is not None:
        t = torch.tensor([elapsed], dtype=torch.float64)
        if use_cuda:
            t = t.cuda()
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if args.breakdown and rank == 0:
        total = sum(peer.phase_times.values())
        print(
            "phase breakdown (%.3fs measured / %.3fs elapsed):" % (total, elapsed),
            file=sys.stderr,
        )
        for k, v in sorted(peer.phase_times.items(), key=lambda kv: -kv[1]):
            print("  %-16s %8.3fs  %5.1f%%" % (k, v, 100 * v / elapsed), file=sys.stderr)

    if rank == 0:
        dtype = "bf16" if (peer.autocast or getattr(peer, "bf16_shadow", False)) else "fp32"
        result = {
            "metric": "env frames/sec (whole node) IMPALA Atari-ResNet",
            "value": frames / elapsed,
            "unit": "frames/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": dtype,
            "data": "synthetic",
            "config": {
                "model": "IMPALA Atari deep ResNet (16/32/32ch, FC3872-256%s)"
                % ("+LSTM256" if args.use_lstm else ""),
                "global_batch": args.batch_size * world,
                "seq_len": args.unroll_length,
                "parallelism": "dp%d" % world,
                "actor_batch_size": args.actor_batch_size,
                "num_actor_batches": args.num_actor_batches,
                "num_actor_cpus": args.num_actor_cpus,
                "virtual_batch_size": args.virtual_batch_size,
                "num_actions": num_actions,
                "frame": "84x84x4 uint8",
            },
        }
        print(json.dumps(result), flush=True)

    if dist is not None:
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    sys.exit(main())
