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

def test_flatten_master_shadow_cpu():
    """Flat-buffer shadow layout (bf16 learner precision mode): params keep
    their values, master<->shadow casts are single flat copies, and
    autograd accumulates into the preassigned flat grad views."""
    import copy

    import torch

    from moolib_amd.impala import flatten_master_shadow

    torch.manual_seed(11)
    master = torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.ReLU(), torch.nn.Linear(16, 4)
    )
    shadow = copy.deepcopy(master).to(torch.bfloat16)
    mp = [p for p in master.parameters() if p.requires_grad]
    sp = [p for p in shadow.parameters() if p.requires_grad]
    before = [p.detach().clone() for p in mp]

    fm, fs, g32, g16 = flatten_master_shadow(mp, sp)

    # values preserved; storage re-homed into the flat buffers
    for p, b in zip(mp, before):
        assert torch.equal(p.detach(), b)
        assert p.data_ptr() >= fm.data_ptr()
        assert p.grad is not None and p.grad.shape == p.shape
    for p in sp:
        assert p.data_ptr() >= fs.data_ptr()

    # optimizer step on master params mutates the flat buffer in place
    opt = torch.optim.SGD(mp, lr=0.1)
    x = torch.randn(3, 8)
    master(x).sum().backward()
    assert g32.abs().sum() > 0  # autograd landed in the flat fp32 buffer
    opt.step()

    # one-shot sync: flat cast copy == per-param fp32->bf16 cast
    fs.copy_(fm)
    for pm, pf in zip(mp, sp):
        assert torch.equal(pf.detach(), pm.detach().to(torch.bfloat16))

    # shadow backward accumulates into the flat bf16 buffer; zeroing the
    # flat buffer zeroes every grad view
    shadow(x.to(torch.bfloat16)).float().sum().backward()
    assert g16.abs().sum() > 0
    assert all(p.grad.abs().sum() > 0 for p in sp)
    g16.zero_()
    assert all(p.grad.abs().sum() == 0 for p in sp)

    # grad cast path: flat bf16 -> flat fp32 equals per-param casts
    shadow(x.to(torch.bfloat16)).float().sum().backward()
    g32.copy_(g16)
    for pm, pf in zip(mp, sp):
        assert torch.equal(pm.grad, pf.grad.float())
