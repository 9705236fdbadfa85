#!/usr/bin/env python
"""Print per-optimizer-step wall times for the first N steps (ramp shape)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import moolib_amd
from moolib_amd.envs import SyntheticAtariEnv
from moolib_amd.impala import ImpalaConfig, ImpalaPeer


def main():
    broker_rpc = moolib_amd.Rpc()
    broker_rpc.set_name("broker")
    broker = moolib_amd.Broker(broker_rpc)
    addr = broker_rpc.listen("127.0.0.1:0")[0]
    cfg = ImpalaConfig(
        num_actions=18, actor_batch_size=128, num_actor_batches=2,
        num_actor_cpus=10, batch_size=32, unroll_length=20,
        virtual_batch_size=32, device="cuda:0", connect=addr,
        total_steps=1e9, lr_schedule=False,
    )
    peer = ImpalaPeer(cfg, lambda: SyntheticAtariEnv(num_actions=18), broker=broker)
    times = []
    events = {"act": 0, "learn": 0, "throttle": 0}
    t_last = None
    while len(times) < 45:
        ev = peer.step_once()
        if ev in events:
            events[ev] += 1
        if ev == "optimize":
            torch.cuda.synchronize()
            t = time.perf_counter()
            if t_last is not None:
                times.append((t - t_last) * 1000)
            t_last = t
            if len(times) % 5 == 0 and times:
                print("steps %2d..%2d: %s  (acts=%d learns=%d throttle=%d)" % (
                    len(times) - 4, len(times),
                    " ".join("%.1f" % x for x in times[-5:]),
                    events["act"], events["learn"], events["throttle"]),
                    flush=True)
                events = {"act": 0, "learn": 0, "throttle": 0}


if __name__ == "__main__":
    main()
