"""Prioritized replay buffer: local semantics + served over RPC."""
import torch

import moolib_amd
from moolib_amd.replay import ReplayBuffer, ReplayClient


class TestReplayLocal:
    def test_ring_and_sample(self):
        buf = ReplayBuffer(capacity=8, alpha=1.0, beta=0.0)
        for i in range(12):  # wraps around
            buf.add({"obs": torch.full((5, 3), float(i)), "act": torch.tensor([i])})
        assert len(buf) == 8
        batch, idx, w = buf.sample(16)
        assert batch["obs"].shape == (16, 5, 3)
        assert batch["act"].shape == (16, 1)
        # ring: values 4..11 survive
        vals = batch["obs"][:, 0, 0]
        assert vals.min() >= 4 and vals.max() <= 11
        assert torch.all(w > 0)

    def test_prioritization_bias(self):
        buf = ReplayBuffer(capacity=4, alpha=1.0, beta=0.5)
        for i in range(4):
            buf.add({"x": torch.tensor([float(i)])}, priority=0.001)
        buf.update_priorities([2], [1000.0])
        batch, idx, w = buf.sample(256)
        frac = (idx == 2).float().mean().item()
        assert frac > 0.95, frac
        # high-priority samples get the smallest importance weight
        assert w[idx == 2].max() <= w.max()

    def test_update_priorities_roundtrip(self):
        buf = ReplayBuffer(capacity=4, alpha=1.0)
        for i in range(4):
            buf.add({"x": torch.tensor([float(i)])}, priority=1.0)
        buf.update_priorities(torch.tensor([0, 1]), torch.tensor([5.0, 0.01]))
        _, idx, _ = buf.sample(200)
        assert (idx == 0).sum() > (idx == 1).sum()


class TestReplayServed:
    def test_rpc_round_trip(self):
        server = moolib_amd.Rpc()
        server.set_name("replay_server")
        addr = server.listen("127.0.0.1:0")[0]
        buf = ReplayBuffer(capacity=16).serve(server, "replay")

        client_rpc = moolib_amd.Rpc()
        client_rpc.set_name("learner")
        client_rpc.set_timeout(15)
        client_rpc.connect(addr)
        client = ReplayClient(client_rpc, "replay_server", "replay")

        futs = [
            client.add({"obs": torch.full((4, 2), float(i)), "r": torch.tensor([0.5 * i])})
            for i in range(10)
        ]
        for f in futs:
            f.result()
        assert client.info()["size"] == 10

        batch, idx, w = client.sample(6).result()
        assert batch["obs"].shape == (6, 4, 2)
        assert w.shape == (6,)
        # priorities update across the wire
        client.update_priorities(idx, torch.ones(6) * 3.0).result()
        assert len(buf) == 10

    def test_sample_ipc_endpoint_cpu(self):
        # On a CPU buffer .sample_ipc degrades to the .cpu() wire path, so
        # clients can use one endpoint everywhere.
        server = moolib_amd.Rpc()
        server.set_name("replay_server2")
        addr = server.listen("127.0.0.1:0")[0]
        buf = ReplayBuffer(capacity=8).serve(server, "replay")

        client_rpc = moolib_amd.Rpc()
        client_rpc.set_name("learner2")
        client_rpc.set_timeout(15)
        client_rpc.connect(addr)
        client = ReplayClient(client_rpc, "replay_server2", "replay")
        for i in range(4):
            client.add({"obs": torch.full((3,), float(i))}).result()
        batch, idx, w = client.sample_ipc(5).result()
        assert batch["obs"].shape == (5, 3)
        assert not batch["obs"].is_cuda
