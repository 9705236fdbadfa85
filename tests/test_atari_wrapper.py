"""AtariPreprocessing wrapper (reference examples/atari/environment.py
parity) tested against a fake ALE-like env — no ROMs needed."""
import numpy as np

from moolib_amd.envs.atari import AtariPreprocessing, resize_area, rgb_to_gray


class FakeALE:
    """210x160x3 env with lives, FIRE action, and frame flicker."""

    def __init__(self, episode_len=100, lives=3):
        self.t = 0
        self.episode_len = episode_len
        self.start_lives = lives
        self.lives = lives
        self.fired = 0
        self.actions = []

    def get_action_meanings(self):
        return ["NOOP", "FIRE", "LEFT", "RIGHT"]

    def _frame(self):
        f = np.zeros((210, 160, 3), dtype=np.uint8)
        # flickering sprite: visible only on even ticks
        if self.t % 2 == 0:
            f[10:20, 10:20] = 255
        f[0, 0] = self.t % 256
        return f

    def reset(self):
        self.t = 0
        self.lives = self.start_lives
        return self._frame()

    def step(self, action):
        self.t += 1
        self.actions.append(action)
        if action == 1:
            self.fired += 1
        if self.t % 40 == 0 and self.lives > 0:
            self.lives -= 1
        done = self.t >= self.episode_len or self.lives == 0
        reward = 3.5 if self.t % 7 == 0 else 0.0
        return self._frame(), reward, done, {"lives": self.lives}


class TestPrimitives:
    def test_gray_matches_luma(self):
        f = np.zeros((4, 4, 3), dtype=np.uint8)
        f[..., 0] = 100  # pure red
        g = rgb_to_gray(f)
        assert g.dtype == np.uint8 and g.shape == (4, 4)
        assert abs(int(g[0, 0]) - int(0.299 * 100)) <= 1

    def test_resize_integer_box_mean(self):
        img = np.arange(16, dtype=np.uint8).reshape(4, 4)
        out = resize_area(img, 2, 2)
        want = img.reshape(2, 2, 2, 2).mean(axis=(1, 3))
        assert np.abs(out.astype(float) - want).max() <= 0.5

    def test_resize_constant_invariant(self):
        img = np.full((210, 160), 77, dtype=np.uint8)
        out = resize_area(img, 84, 84)
        assert out.shape == (84, 84)
        assert np.all(out == 77)


class TestWrapper:
    def test_obs_shape_and_dtype(self):
        env = AtariPreprocessing(FakeALE(), noop_max=0)
        obs = env.reset()
        assert obs.shape == (4, 84, 84) and obs.dtype == np.uint8
        obs, r, done, info = env.step(0)
        assert obs.shape == (4, 84, 84)

    def test_frame_stack_rolls(self):
        env = AtariPreprocessing(FakeALE(), noop_max=0)
        env.reset()
        o1, *_ = env.step(0)
        o2, *_ = env.step(0)
        # newest frame is at index -1; previous newest shifted to -2
        assert np.array_equal(o2[-2], o1[-1])

    def test_reward_clipping(self):
        env = AtariPreprocessing(FakeALE(), noop_max=0, frame_skip=7)
        env.reset()
        _, r, _, _ = env.step(0)
        assert r == 1.0  # 3.5 clipped
        env2 = AtariPreprocessing(FakeALE(), noop_max=0, frame_skip=7, clip_rewards=False)
        env2.reset()
        _, r2, _, _ = env2.step(0)
        assert r2 == 3.5

    def test_fire_reset_and_noops(self):
        inner = FakeALE()
        env = AtariPreprocessing(inner, noop_max=5, rng=np.random.RandomState(0))
        env.reset()
        assert inner.fired >= 1  # FIRE pressed on reset
        assert all(a in (0, 1) for a in inner.actions)

    def test_episodic_life(self):
        inner = FakeALE(episode_len=1000)
        env = AtariPreprocessing(inner, noop_max=0)
        env.reset()
        # lives drop at t=40 -> with frame_skip 4 that's step 10
        saw_life_done = False
        for _ in range(15):
            _, _, done, info = env.step(0)
            if done:
                saw_life_done = True
                break
        assert saw_life_done
        assert not env._real_done  # game not actually over
        env.reset()  # soft reset: game continues
        assert inner.t > 40  # env was NOT hard reset

    def test_flicker_max(self):
        env = AtariPreprocessing(FakeALE(), noop_max=0)
        env.reset()
        obs, *_ = env.step(0)
        # the sprite flickers on odd ticks, but max over the last two
        # skip frames keeps it visible in the processed frame
        assert obs[-1][4:8, 4:8].max() > 100

    def test_envpool_integration(self):
        import torch

        import moolib_amd

        pool = moolib_amd.EnvPool(
            lambda: AtariPreprocessing(FakeALE(), noop_max=0),
            num_processes=2,
            batch_size=4,
            num_batches=1,
        )
        obs = pool.step(0, torch.zeros(4, dtype=torch.int64)).result()
        frame = obs["state"] if isinstance(obs, dict) and "state" in obs else obs
        import moolib_amd.utils.nest as nest

        flat = nest.flatten(frame)
        shapes = [tuple(t.shape) for t in flat]
        assert any(s[-3:] == (4, 84, 84) for s in shapes), shapes
        del pool
