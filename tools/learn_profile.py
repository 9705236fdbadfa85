#!/usr/bin/env python
"""Profile ONE learner step (compute_gradients) by torch op, to attribute
the eager elementwise soup the kernel-level profile can't name.

Run on an MI355X:  python tools/learn_profile.py
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
        num_actions=18,
        actor_batch_size=128,
        num_actor_batches=2,
        num_actor_cpus=10,
        batch_size=32,
        unroll_length=20,
        virtual_batch_size=32,
        device="cuda:0",
        connect=addr,
        total_steps=1e9,
    )
    peer = ImpalaPeer(cfg, lambda: SyntheticAtariEnv(num_actions=18), broker=broker)
    # warm up until a few optimizer steps have happened (find settles, graphs capture)
    done = 0
    while done < 6:
        if peer.step_once() == "optimize":
            done += 1
    # grab one learn batch and profile compute_gradients alone
    while peer.learn_batcher.empty():
        peer.step_once()
    data = peer.learn_batcher.get()
    torch.cuda.synchronize()
    from torch.profiler import ProfilerActivity, profile

    with profile(activities=[ProfilerActivity.CUDA], record_shapes=True) as prof:
        for _ in range(3):
            peer.compute_gradients(data)
        torch.cuda.synchronize()
    print(prof.key_averages(group_by_input_shape=False).table(
        sort_by="self_cuda_time_total", row_limit=35))

    # second pass WITH CPU activity: attributes CUDA time to aten ops (and
    # their shapes), which names the generic elementwise kernels
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=True) as prof2:
        for _ in range(3):
            peer.compute_gradients(data)
        torch.cuda.synchronize()
    print(prof2.key_averages(group_by_input_shape=True).table(
        sort_by="cuda_time_total", row_limit=45))


if __name__ == "__main__":
    main()
