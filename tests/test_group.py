"""Broker / Group / AllReduce tests (multi-peer in one process).

Mirrors the coverage of the reference's test/test_group.py and
test/test_reduce.py (membership, sort order, peer churn, allreduce
consistency) against our implementation.
"""
import time

import torch

import moolib_amd


class Cluster:
    def __init__(self, n, group="g", timeout=3.0):
        self.broker_rpc = moolib_amd.Rpc()
        self.broker_rpc.set_name("broker")
        self.broker = moolib_amd.Broker(self.broker_rpc)
        self.addr = self.broker_rpc.listen("127.0.0.1:0")[0]
        self.peers = []
        self.groups = []
        for i in range(n):
            self.add_peer(i, group, timeout)

    def add_peer(self, i, group="g", timeout=3.0):
        r = moolib_amd.Rpc()
        r.set_name("peer%d" % i)
        r.set_timeout(timeout)
        r.connect(self.addr)
        g = moolib_amd.Group(r, group)
        g.set_timeout(timeout)
        self.peers.append(r)
        self.groups.append(g)
        return g

    def pump(self, cond, deadline=20):
        t0 = time.time()
        while time.time() - t0 < deadline:
            self.broker.update()
            for g in self.groups:
                g.update()
            if cond():
                return True
            time.sleep(0.02)
        return False

    def wait_active(self, n=None):
        n = n if n is not None else len(self.groups)
        ok = self.pump(
            lambda: all(g.active() for g in self.groups)
            and all(len(g.members()) == n for g in self.groups)
        )
        assert ok, "group formation failed: %s" % [(g.active(), g.members()) for g in self.groups]

    def reduce_all(self, name, values, op=None):
        futs = [g.all_reduce(name, v, op) for g, v in zip(self.groups, values)]
        ok = self.pump(lambda: all(f.done() for f in futs))
        assert ok, "allreduce did not complete"
        return [f.result() for f in futs]


class TestGroup:
    def test_membership_and_sync(self):
        c = Cluster(4)
        c.wait_active()
        ms = c.groups[0].members()
        assert sorted(ms) == ["peer0", "peer1", "peer2", "peer3"]
        assert len(set(g.sync_id() for g in c.groups)) == 1

    def test_sort_order(self):
        c = Cluster(0)
        for i in range(3):
            g = c.add_peer(i)
        for i, g in enumerate(c.groups):
            g.set_sort_order(100 - i * 10)  # reverse order
        c.wait_active()
        # sort order dominates join order
        deadline = time.time() + 10
        while time.time() < deadline:
            if c.groups[0].members() == ["peer2", "peer1", "peer0"]:
                break
            c.pump(lambda: False, deadline=0.3)
        assert c.groups[0].members() == ["peer2", "peer1", "peer0"]

    def test_allreduce_sum_int(self):
        c = Cluster(5)
        c.wait_active()
        res = c.reduce_all("s", [i for i in range(5)])
        assert res == [10] * 5

    def test_allreduce_tensor(self):
        c = Cluster(3)
        c.wait_active()
        vals = [torch.randn(17, 5) for _ in range(3)]
        want = vals[0] + vals[1] + vals[2]
        res = c.reduce_all("t", vals)
        for r in res:
            assert torch.allclose(r, want, atol=1e-5)

    def test_allreduce_custom_op(self):
        c = Cluster(4)
        c.wait_active()

        def fold(a, b):
            for k, v in b.items():
                a[k] = max(a[k], v)
            return a

        res = c.reduce_all("m", [{"x": i * 2} for i in range(4)], fold)
        assert all(r["x"] == 6 for r in res)

    def test_repeated_allreduce(self):
        c = Cluster(3)
        c.wait_active()
        for round_i in range(5):
            res = c.reduce_all("r", [round_i * 10 + i for i in range(3)])
            assert res == [round_i * 30 + 3] * 3

    def test_member_leaves(self):
        c = Cluster(3, timeout=2.0)
        c.wait_active()
        old_sync = c.groups[0].sync_id()
        # Kill peer2
        dead_group = c.groups.pop()
        dead_peer = c.peers.pop()
        del dead_group, dead_peer
        ok = c.pump(
            lambda: all(len(g.members()) == 2 for g in c.groups)
            and all(g.sync_id() != old_sync for g in c.groups),
            deadline=15,
        )
        assert ok, "eviction did not propagate"
        # allreduce still works with the survivors
        res = c.reduce_all("after", [1, 2])
        assert res == [3, 3]

    def test_late_joiner(self):
        c = Cluster(2)
        c.wait_active()
        c.add_peer(2)
        c.wait_active(3)
        res = c.reduce_all("late", [1, 1, 1])
        assert res == [3, 3, 3]

    def test_allreduce_fails_without_group(self):
        r = moolib_amd.Rpc()
        r.set_name("solo")
        g = moolib_amd.Group(r, "nope")
        fut = g.all_reduce("x", 1)
        fut.wait(10)
        assert isinstance(fut.exception(), moolib_amd.RpcError)


class TestCombinedBrokerMember:
    def test_one_rpc_is_broker_and_member(self):
        """Self-call dispatch lets a single Rpc host the Broker AND join a
        group (pings to 'broker' == itself resolve locally)."""
        rpc = moolib_amd.Rpc()
        rpc.set_name("broker")
        broker = moolib_amd.Broker(rpc)
        rpc.listen("127.0.0.1:0")
        g = moolib_amd.Group(rpc, "combined")
        g.set_timeout(5)
        t0 = time.time()
        while time.time() - t0 < 15 and not g.active():
            broker.update()
            g.update()
            time.sleep(0.02)
        assert g.active() and g.members() == ["broker"]
        fut = g.all_reduce("x", 7)
        t0 = time.time()
        while time.time() - t0 < 10 and not fut.done():
            broker.update()
            g.update()
            time.sleep(0.02)
        assert fut.result() == 7


class TestWideGroup:
    def test_twelve_peer_tree_allreduce(self):
        """Deeper binary tree (12 members, 4 levels) with churn-free sums."""
        c = Cluster(12)
        c.wait_active()
        for r in range(3):
            res = c.reduce_all("wide", [r * 100 + i for i in range(12)])
            assert res == [r * 1200 + 66] * 12
