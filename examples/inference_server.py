"""Dynamically-batched inference server over the RPC plane.

The reference's `define(batch_size=..., dynamic_batching=True)` pattern
(src/moolib.cc:1007-1178, test/test_batch.py): many actors submit single
observations; the server stacks them into batches for the model, and the
latency model (moolib_amd/__init__.py _BatchCollector) decides when a
partial batch is worth flushing. On MI355X the handler runs the AtariNet
forward in bf16 (channels_last, fused kernels when built).

Run:  python examples/inference_server.py [--device cuda:0] [--clients 8]
"""
import argparse
import threading
import time

import torch

import moolib_amd
from moolib_amd.models.atari import AtariNet


def run(device=None, clients=8, batch_size=32, seconds=10.0, num_actions=18):
    """Returns (total_requests, batch_sizes); used by tests and main()."""
    device = device or ("cuda:0" if torch.cuda.is_available() else "cpu")

    class args:
        pass

    args.clients, args.batch_size = clients, batch_size
    args.seconds, args.num_actions = seconds, num_actions

    model = AtariNet(num_actions=args.num_actions).to(device)
    if device.startswith("cuda"):
        model = model.to(memory_format=torch.channels_last)
    model.eval()

    server = moolib_amd.Rpc()
    server.set_name("inference")
    addr = [a for a in server.listen("127.0.0.1:0") if a.startswith("tcp://127")][0]

    batch_sizes = []

    def act(state, reward, prev_action):
        # batched: state [B, 4, 84, 84] uint8 on `device`
        B = state.shape[0]
        batch_sizes.append(B)
        inputs = {
            "state": state.unsqueeze(0),  # [T=1, B, ...]
            "reward": reward.view(1, B),
            "prev_action": prev_action.view(1, B),
            "done": torch.zeros(1, B, dtype=torch.bool, device=state.device),
        }
        with torch.no_grad():
            out, _ = model(inputs, tuple())
        return out["action"].view(B)  # split back per caller by the collector

    server.define(
        "act", act, batch_size=args.batch_size, device=device, dynamic_batching=True
    )

    done = threading.Event()
    counts = [0] * args.clients

    def client(i):
        rpc = moolib_amd.Rpc()
        rpc.set_name("client%d" % i)
        rpc.set_timeout(30)
        rpc.connect(addr)
        frame = torch.randint(0, 255, (4, 84, 84), dtype=torch.uint8)
        action = torch.tensor(0, dtype=torch.int64)
        while not done.is_set():
            action = rpc.sync(
                "inference", "act", frame, torch.tensor(0.0), action
            )
            counts[i] += 1

    threads = [threading.Thread(target=client, args=(i,), daemon=True) for i in range(args.clients)]
    t0 = time.time()
    for t in threads:
        t.start()
    time.sleep(args.seconds)
    done.set()
    for t in threads:
        t.join(timeout=10)
    dt = time.time() - t0
    total = sum(counts)
    mean_b = sum(batch_sizes) / max(len(batch_sizes), 1)
    print(
        "inference server: %d requests in %.1fs (%.0f req/s), %d batches "
        "(mean batch %.1f, max %d) on %s"
        % (total, dt, total / dt, len(batch_sizes), mean_b, max(batch_sizes or [0]), device)
    )
    return total, batch_sizes


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default=None)
    ap.add_argument("--clients", type=int, default=8)
    ap.add_argument("--batch-size", type=int, default=32)
    ap.add_argument("--seconds", type=float, default=10.0)
    ap.add_argument("--num-actions", type=int, default=18)
    a = ap.parse_args()
    run(a.device, a.clients, a.batch_size, a.seconds, a.num_actions)


if __name__ == "__main__":
    main()
