"""EnvPool tests with a deterministic toy env (no gym dependency).

Mirrors the reference's test/unit/test_envpool.py strategy: validate
double-buffered stepping semantics, auto-reset behavior, and field shapes.
"""
import time

import numpy as np
import pytest
import torch

import moolib_amd


class CountingEnv:
    """obs = [count, last_action]; episode ends after 5 steps; reward = action."""

    def __init__(self):
        self.count = 0

    def reset(self):
        self.count = 0
        return np.array([self.count, -1.0], dtype=np.float32)

    def step(self, action):
        self.count += 1
        done = self.count >= 5
        obs = np.array([self.count, float(action)], dtype=np.float32)
        reward = float(action)
        if done:
            # the pool auto-resets and returns the new episode's first obs
            pass
        return obs, reward, done, {}


class DictObsEnv:
    def __init__(self):
        self.t = 0

    def reset(self):
        self.t = 0
        return {"a": np.zeros((2, 3), dtype=np.float32), "b": np.int64(self.t)}

    def step(self, action):
        self.t += 1
        obs = {"a": np.full((2, 3), float(self.t), dtype=np.float32), "b": np.int64(self.t)}
        return obs, 1.0, self.t >= 3, {}


class TestEnvPool:
    def test_basic_stepping(self):
        pool = moolib_amd.EnvPool(CountingEnv, num_processes=2, batch_size=4, num_batches=1)
        actions = torch.zeros(4, dtype=torch.int64)
        # First step: reset -> count 0, reward 0, done False
        obs = pool.step(0, actions).result()
        assert set(obs.keys()) == {"state", "reward", "done"}
        assert obs["state"].shape == (4, 2)
        assert torch.equal(obs["state"][:, 0], torch.zeros(4))
        assert torch.equal(obs["reward"], torch.zeros(4))
        assert not obs["done"].any()
        # Steps 1..4: counting up, reward = action
        for i in range(1, 5):
            actions = torch.full((4,), i % 3, dtype=torch.int64)
            obs = pool.step(0, actions).result()
            if i < 5:
                assert torch.equal(obs["state"][:, 0], torch.full((4,), float(i)))
                assert torch.equal(obs["state"][:, 1], torch.full((4,), float(i % 3)))
                assert torch.allclose(obs["reward"], torch.full((4,), float(i % 3)))
                assert not obs["done"].any()
        # Step 5: done; obs is the auto-reset initial obs
        obs = pool.step(0, torch.ones(4, dtype=torch.int64)).result()
        assert obs["done"].all()
        assert torch.equal(obs["state"][:, 0], torch.zeros(4))
        assert torch.allclose(obs["reward"], torch.ones(4))
        # After reset: counting starts again
        obs = pool.step(0, torch.zeros(4, dtype=torch.int64)).result()
        assert not obs["done"].any()
        assert torch.equal(obs["state"][:, 0], torch.ones(4))

    def test_double_buffering(self):
        pool = moolib_amd.EnvPool(CountingEnv, num_processes=2, batch_size=3, num_batches=2)
        f0 = pool.step(0, torch.zeros(3, dtype=torch.int64))
        f1 = pool.step(1, torch.zeros(3, dtype=torch.int64))
        o0, o1 = f0.result(), f1.result()
        assert torch.equal(o0["state"][:, 0], torch.zeros(3))
        assert torch.equal(o1["state"][:, 0], torch.zeros(3))
        # batches advance independently
        pool.step(0, torch.zeros(3, dtype=torch.int64)).result()
        o0b = pool.step(0, torch.zeros(3, dtype=torch.int64)).result()
        o1b = pool.step(1, torch.zeros(3, dtype=torch.int64)).result()
        assert torch.equal(o0b["state"][:, 0], torch.full((3,), 2.0))
        assert torch.equal(o1b["state"][:, 0], torch.full((3,), 1.0))

    def test_dict_obs(self):
        pool = moolib_amd.EnvPool(DictObsEnv, num_processes=1, batch_size=2, num_batches=1)
        obs = pool.step(0, torch.zeros(2, dtype=torch.int64)).result()
        assert set(obs.keys()) == {"a", "b", "reward", "done"}
        assert obs["a"].shape == (2, 2, 3)
        assert obs["a"].dtype == torch.float32
        assert obs["b"].dtype == torch.int64
        obs = pool.step(0, torch.zeros(2, dtype=torch.int64)).result()
        assert torch.allclose(obs["a"], torch.ones(2, 2, 3))
        assert torch.equal(obs["b"], torch.ones(2, dtype=torch.int64))

    def test_worker_error_surfaces(self):
        class BadEnv:
            def reset(self):
                raise ValueError("bad env is bad")

            def step(self, a):
                return None

        pool = moolib_amd.EnvPool(BadEnv, num_processes=1, batch_size=1, num_batches=1)
        with pytest.raises(RuntimeError, match="bad env is bad|died|failed"):
            pool.step(0, torch.zeros(1, dtype=torch.int64)).result()

    def test_running(self):
        pool = moolib_amd.EnvPool(CountingEnv, num_processes=2, batch_size=2, num_batches=1)
        assert pool.running()
        assert pool.num_workers_alive() == 2


class FdProbeEnv:
    """Observation = number of open socket fds in the worker process."""

    def reset(self):
        return self._probe()

    def step(self, action):
        return self._probe(), 0.0, False, {}

    def _probe(self):
        import os

        n = 0
        for fd in os.listdir("/proc/self/fd"):
            try:
                if "socket:" in os.readlink("/proc/self/fd/" + fd):
                    n += 1
            except OSError:
                pass
        return np.array([n], dtype=np.int64)


class TestForkHygiene:
    def _worker_socket_count(self):
        import moolib_amd.utils.nest as nest

        pool = moolib_amd.EnvPool(FdProbeEnv, num_processes=1, batch_size=2, num_batches=1)
        obs = pool.step(0, torch.zeros(2, dtype=torch.int64)).result()
        n = int(list(nest.flatten(obs))[0].max())
        del pool
        return n

    def test_workers_inherit_no_engine_sockets(self):
        # A live RPC plane in the parent must not leak its sockets into
        # EnvPool workers (pthread_atfork handler in csrc/socket.cc).
        # Differential: ambient fds from other test machinery may exist, so
        # compare worker socket counts before/after standing up an RPC plane
        # with live connections.
        base = self._worker_socket_count()
        host = moolib_amd.Rpc()
        host.set_name("fd_host")
        addr = host.listen("127.0.0.1:0")[0]
        client = moolib_amd.Rpc()
        client.set_name("fd_client")
        client.set_timeout(15)
        client.connect(addr)
        host.define("hi", lambda: 1)
        assert client.sync("fd_host", "hi") == 1  # live conns on both ends
        after = self._worker_socket_count()
        assert after <= base, (base, after)


class FailsAtStep3Env:
    def __init__(self):
        self.n = 0

    def reset(self):
        return np.zeros(2, dtype=np.float32)

    def step(self, action):
        self.n += 1
        if self.n == 3:
            raise RuntimeError("env exploded at step 3")
        return np.zeros(2, dtype=np.float32), 1.0, False, {}


class TestMidEpisodeError:
    def test_step_error_surfaces_not_hangs(self):
        pool = moolib_amd.EnvPool(FailsAtStep3Env, num_processes=1, batch_size=2, num_batches=1)
        import pytest as _pt

        with _pt.raises(RuntimeError, match="exploded|died|failed"):
            for i in range(10):
                pool.step(0, torch.zeros(2, dtype=torch.int64)).result()


def _runner_proc(shm_name, stop_file):
    import os
    import time as _t

    import moolib_amd as M

    runner = M.EnvRunner(CountingEnv)
    runner.start(shm_name)
    t0 = _t.time()
    while not os.path.exists(stop_file) and _t.time() - t0 < 120:
        _t.sleep(0.05)


class TestExternalEnvRunner:
    def test_runners_in_separate_processes(self, tmp_path):
        """EnvPool(shm_name=..., external_workers=True) is served by
        EnvRunner processes launched independently (reference env-server
        role, src/env.h:363-453)."""
        import multiprocessing as mp
        import uuid

        name = "t" + uuid.uuid4().hex[:10]
        stop_file = str(tmp_path / "stop")
        ctx = mp.get_context("spawn")
        procs = [
            ctx.Process(target=_runner_proc, args=(name, stop_file)) for _ in range(2)
        ]
        for p in procs:
            p.start()
        try:
            pool = moolib_amd.EnvPool(
                CountingEnv,
                num_processes=2,
                batch_size=4,
                num_batches=2,
                shm_name=name,
                external_workers=True,
            )
            actions = torch.arange(4, dtype=torch.int64)
            obs = pool.step(0, torch.zeros(4, dtype=torch.int64)).result()
            assert torch.equal(obs["state"][:, 0], torch.zeros(4))
            for i in range(1, 5):
                obs = pool.step(0, actions).result()
                assert torch.equal(obs["state"][:, 0], torch.full((4,), float(i)))
                assert torch.equal(obs["reward"], actions.float())
            obs = pool.step(0, actions).result()  # 5th env step ends the episode
            assert obs["done"].all()
            # the second buffer works too
            obs1 = pool.step(1, torch.zeros(4, dtype=torch.int64)).result()
            assert torch.equal(obs1["state"][:, 0], torch.zeros(4))
            assert pool.num_workers_alive() == 2
            del pool
        finally:
            open(stop_file, "w").write("x")
            for p in procs:
                p.join(timeout=30)
                if p.is_alive():
                    p.kill()

    def test_slot_overclaim_rejected(self, tmp_path):
        import uuid

        name = "t" + uuid.uuid4().hex[:10]
        pool = moolib_amd.EnvPool(
            CountingEnv, num_processes=1, batch_size=2, num_batches=1,
            shm_name=name, external_workers=True,
        )
        r1 = moolib_amd.EnvRunner(CountingEnv)
        r1.start(name)
        r2 = moolib_amd.EnvRunner(CountingEnv)
        with pytest.raises(Exception, match="slots claimed"):
            r2.start(name)
        obs = pool.step(0, torch.zeros(2, dtype=torch.int64)).result()
        assert obs["state"].shape == (2, 2)
        assert r1.running()
        del pool


class TestPoll:
    def test_poll_semantics(self):
        """poll() is a non-consuming completion check: false before the
        step completes/after result() consumed it, true in between."""
        pool = moolib_amd.EnvPool(CountingEnv, num_processes=2, batch_size=4, num_batches=1)
        assert not pool.poll(0)  # nothing stepped yet
        fut = pool.step(0, torch.zeros(4, dtype=torch.int64))
        t0 = time.time()
        while not pool.poll(0) and time.time() - t0 < 30:
            time.sleep(0.002)
        assert pool.poll(0)      # complete, not yet consumed
        assert pool.poll(0)      # non-consuming: still true
        fut.result()
        assert not pool.poll(0)  # consumed
        assert not pool.poll(7)  # out of range
