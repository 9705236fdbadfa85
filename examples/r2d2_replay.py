"""R2D2-style distributed prioritized replay demo (BASELINE config 5).

One process hosts an HBM-resident prioritized ReplayBuffer served over the
moolib RPC plane; an actor peer streams LSTM rollout sequences into it; a
learner peer samples prioritized batches and updates priorities. On one
node, `--ipc` serves samples as hipIpc handles (zero-copy GPU->GPU).

Run:  python examples/r2d2_replay.py [--device cuda:0] [--ipc] [--seconds 20]
"""
import argparse
import time

import torch

import moolib_amd
from moolib_amd.envs import SyntheticAtariEnv
from moolib_amd.models.atari import AtariNet
from moolib_amd.replay import ReplayBuffer
from moolib_amd.utils import nest


def run(device=None, capacity=2048, unroll=40, batch_size=16, num_envs=32,
        seconds=20.0, ipc=False):
    """Run the replay workload; returns a metrics dict (also used by
    `bench.py --config r2d2`)."""
    device = device or ("cuda:0" if torch.cuda.is_available() else "cpu")

    class args:  # keep the body below unchanged
        pass

    args.capacity, args.unroll, args.batch_size = capacity, unroll, batch_size
    args.num_envs, args.seconds, args.ipc = num_envs, seconds, ipc

    # Replay server peer: sequences live in HBM.
    server_rpc = moolib_amd.Rpc()
    server_rpc.set_name("replay_server")
    addr = [a for a in server_rpc.listen("127.0.0.1:0") if a.startswith("tcp://127")][0]
    buf = ReplayBuffer(args.capacity, device=device, alpha=0.6, beta=0.4)
    buf.serve(server_rpc, "replay")  # serves .sample and .sample_ipc

    # Actor: generate LSTM rollouts from synthetic frames.
    envs = moolib_amd.EnvPool(
        lambda: SyntheticAtariEnv(num_actions=18),
        num_processes=4,
        batch_size=args.num_envs,
        num_batches=1,
    )
    model = AtariNet(num_actions=18, use_lstm=True).to(device)
    actor_rpc = moolib_amd.Rpc()
    actor_rpc.set_name("actor")
    actor_rpc.set_timeout(30)
    actor_rpc.connect(addr)

    learner_rpc = moolib_amd.Rpc()
    learner_rpc.set_name("learner")
    learner_rpc.set_timeout(30)
    learner_rpc.connect(addr)

    time_batcher = moolib_amd.Batcher(args.unroll, device)
    core_state = tuple(s.to(device) for s in model.initial_state(batch_size=args.num_envs))
    prev_action = torch.zeros(args.num_envs, dtype=torch.int64, device=device)

    t0 = time.time()
    added = sampled = 0
    frames = 0
    add_futures = []
    while time.time() - t0 < args.seconds:
        # ---- act ----
        obs = envs.step(0, prev_action).result()
        dev = {k: t.to(device, copy=True) for k, t in obs.items()}
        dev["prev_action"] = prev_action
        with torch.no_grad():
            out, core_state = model(
                nest.map(lambda t: t.unsqueeze(0), dev), core_state
            )
        prev_action = out["action"].squeeze(0)
        frames += args.num_envs
        time_batcher.stack(
            {"state": dev["state"], "action": prev_action, "reward": dev["reward"]}
        )
        # ---- feed replay (sequence-major, one add per unroll) ----
        if not time_batcher.empty():
            seq = time_batcher.get()  # [T, B, ...]
            add_futures.append(
                actor_rpc.async_("replay_server", "replay.add", nest.map(lambda t: t.cpu(), seq))
            )
        add_futures = [f for f in add_futures if not f.done()]
        added = len(buf)

        # ---- learner: prioritized sample + priority update ----
        if len(buf) >= args.batch_size:
            fn = "replay.sample_ipc" if (args.ipc and device.startswith("cuda")) else "replay.sample"
            batch, idx, w = learner_rpc.sync("replay_server", fn, args.batch_size)
            td_error = torch.rand(args.batch_size)  # stand-in for the R2D2 TD error
            learner_rpc.sync("replay_server", "replay.update_priorities", idx, td_error)
            sampled += args.batch_size

    dt = time.time() - t0
    return {
        "seconds": dt,
        "frames_acted": frames,
        "frames_per_s": frames / dt,
        "buffer_seqs": len(buf),
        "capacity": args.capacity,
        "sampled": sampled,
        "sampled_per_s": sampled / dt,
        "unroll": args.unroll,
        "batch_size": args.batch_size,
        "num_envs": args.num_envs,
        "ipc": ipc,
        "device": str(device),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default=None)
    ap.add_argument("--capacity", type=int, default=2048)
    ap.add_argument("--unroll", type=int, default=40)
    ap.add_argument("--batch-size", type=int, default=16)
    ap.add_argument("--num-envs", type=int, default=32)
    ap.add_argument("--seconds", type=float, default=20.0)
    ap.add_argument("--ipc", action="store_true", help="serve samples as hipIpc handles")
    a = ap.parse_args()
    m = run(a.device, a.capacity, a.unroll, a.batch_size, a.num_envs, a.seconds, a.ipc)
    print(
        "r2d2 replay demo: %.1fs, %d frames acted (%.0f/s), buffer %d/%d seqs, "
        "%d sequences sampled (%.0f/s)"
        % (
            m["seconds"], m["frames_acted"], m["frames_per_s"], m["buffer_seqs"],
            m["capacity"], m["sampled"], m["sampled_per_s"],
        )
    )


if __name__ == "__main__":
    main()
