#!/usr/bin/env python
"""Profile FULL steady-state step_once iterations (acts + learns + optimize)
with aten attribution, to name the ops the kernel-level profile can't:
the copyBuffer (D2D memcpy) and fp32-div populations seen in
profiles/r4b_final_kernel_stats.csv live outside compute_gradients.

Run on an MI355X:  python tools/step_profile.py
"""
import os
import sys

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
    done = 0
    while done < 25:  # settle: finds done, graphs captured
        if peer.step_once() == "optimize":
            done += 1
    torch.cuda.synchronize()
    from torch.profiler import ProfilerActivity, profile

    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=True) as prof:
        done = 0
        while done < 5:
            if peer.step_once() == "optimize":
                done += 1
        torch.cuda.synchronize()
    print(prof.key_averages(group_by_input_shape=True).table(
        sort_by="cuda_time_total", row_limit=60))


if __name__ == "__main__":
    main()
