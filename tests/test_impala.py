"""End-to-end IMPALA engine test on CPU with a tiny synthetic config.

Exercises the full cooperative loop: EnvPool acting, time/learn batching,
V-trace gradients, Accumulator reduction, optimizer stepping.
"""
import time

import pytest

import moolib_amd
from moolib_amd.envs import SyntheticAtariEnv
from moolib_amd.impala import ImpalaConfig, ImpalaPeer


def tiny_config(addr):
    return ImpalaConfig(
        num_actions=6,
        actor_batch_size=8,
        num_actor_batches=2,
        num_actor_cpus=2,
        batch_size=4,
        unroll_length=5,
        virtual_batch_size=4,
        device="cpu",
        autocast_bf16=False,
        connect=addr,
        total_steps=1e6,
    )


@pytest.mark.timeout(300)
def test_impala_cpu_end_to_end():
    broker_rpc = moolib_amd.Rpc()
    broker_rpc.set_name("broker")
    broker = moolib_amd.Broker(broker_rpc)
    addr = broker_rpc.listen("127.0.0.1:0")[0]

    cfg = tiny_config(addr)
    peer = ImpalaPeer(
        cfg, lambda: SyntheticAtariEnv(num_actions=6, mean_episode_len=50), broker=broker
    )

    t0 = time.time()
    events = {"optimize": 0, "learn": 0, "act": 0, "idle": 0, "throttle": 0}
    while events["optimize"] < 3 and time.time() - t0 < 200:
        ev = peer.step_once()
        events[ev] += 1

    assert events["optimize"] >= 3, events
    assert events["learn"] >= 3
    # acting may happen opportunistically inside learn/optimize iterations
    # (env-fed scheduling), so count env frames rather than "act" returns
    assert peer.stats["env_train_steps"].result() >= 3 * cfg.unroll_length * cfg.batch_size
    assert peer.stats["optimizer_steps"].result() == events["optimize"]
    # parameters actually moved
    total_norm = sum(p.detach().norm().item() for p in peer.model.parameters())
    assert total_norm == total_norm  # not NaN
    assert peer.model_version == events["optimize"]
