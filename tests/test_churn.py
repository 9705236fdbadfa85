"""Elastic stress: peers join and leave while training continues.

Mirrors the reference's test/test_reduce.py pattern (repeated reductions
with random peer churn and consistency assertions).
"""
import time

import pytest
import torch

import moolib_amd


class ChurnPeer:
    def __init__(self, name, addr):
        self.rpc = moolib_amd.Rpc()
        self.rpc.set_name(name)
        self.rpc.set_timeout(4)
        self.group = moolib_amd.Group(self.rpc, "churn")
        self.group.set_timeout(3)
        torch.manual_seed(hash(name) % 1000)
        self.params = [torch.randn(6, 4).requires_grad_()]
        self.acc = moolib_amd.Accumulator("acc", self.params, [], group=self.group)
        self.acc.set_virtual_batch_size(1)
        self.acc.connect(addr)
        self.steps = 0

    def pump(self):
        self.acc.update()
        if self.acc.wants_state():
            self.acc.set_state({"v": self.acc.model_version()})
        if self.acc.has_new_state():
            self.acc.state()
        if self.acc.connected() and self.acc.wants_gradients():
            self.params[0].grad = torch.ones_like(self.params[0])
            self.acc.reduce_gradients(4)
        if self.acc.has_gradients():
            stats = self.acc.get_gradient_stats()
            n = max(stats["num_gradients"], 1)
            # result must be the average of identical unit gradients
            assert torch.allclose(
                self.params[0].grad, torch.ones_like(self.params[0])
            ), (stats, self.params[0].grad.flatten()[:4])
            with torch.no_grad():
                self.params[0] -= 0.0 * self.params[0].grad
            self.acc.zero_gradients()
            self.steps += 1


@pytest.mark.timeout(300)
def test_training_survives_peer_churn():
    broker_rpc = moolib_amd.Rpc()
    broker_rpc.set_name("broker")
    broker = moolib_amd.Broker(broker_rpc)
    addr = broker_rpc.listen("127.0.0.1:0")[0]

    peers = {i: ChurnPeer("peer%d" % i, addr) for i in range(3)}
    next_id = 3
    t0 = time.time()
    churned = 0
    baseline_steps = 0

    while time.time() - t0 < 90:
        broker.update()
        for p in peers.values():
            p.pump()
        total = sum(p.steps for p in peers.values())
        # churn: every time peer0 makes ~5 more steps, kill one other peer
        # and add a fresh one
        if total - baseline_steps >= 12 and churned < 3:
            baseline_steps = total
            victim = sorted(k for k in peers if k != 0)[0]
            del peers[victim]
            peers[next_id] = ChurnPeer("peer%d" % next_id, addr)
            next_id += 1
            churned += 1
        if churned >= 3 and all(p.steps >= 3 for p in peers.values()):
            break
        time.sleep(0.002)

    assert churned == 3, "churn did not happen (total steps: %d)" % total
    # every surviving peer (including late joiners) made progress after churn
    for k, p in peers.items():
        assert p.steps >= 3, "peer%d stalled (steps=%d)" % (k, p.steps)
    # model versions agree across survivors
    versions = {p.acc.model_version() for p in peers.values()}
    assert len(versions) <= 2, versions  # at most off-by-one mid-round
