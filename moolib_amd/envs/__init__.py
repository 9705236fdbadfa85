"""Built-in environments: a pure-python CartPole (no gym dependency) and a
synthetic Atari-frame env for benchmarking.

The synthetic env produces the exact observation shape the reference's
benchmark config consumes (uint8 [4, 84, 84] stacked frames,
examples/atari/atari_preprocessing.py + FrameStack(4)) without ALE: frames
are drawn from a pre-generated bank, episodes end stochastically. There is
no network access for ROMs/datasets, so the headline benchmark runs on this
("synthetic 84x84x4 frames").
"""
import math

import numpy as np


class CartPoleEnv:
    """Classic cart-pole dynamics (Barto, Sutton & Anderson), pure python.

    Same observation/action/reward/termination spec as gym's CartPole-v1.
    """

    def __init__(self, seed=None):
        self.gravity = 9.8
        self.masscart = 1.0
        self.masspole = 0.1
        self.total_mass = self.masspole + self.masscart
        self.length = 0.5
        self.polemass_length = self.masspole * self.length
        self.force_mag = 10.0
        self.tau = 0.02
        self.theta_threshold = 12 * 2 * math.pi / 360
        self.x_threshold = 2.4
        self.max_steps = 500
        self.rng = np.random.RandomState(seed)
        self.state = None
        self.steps = 0

    def reset(self):
        self.state = self.rng.uniform(low=-0.05, high=0.05, size=(4,)).astype(np.float32)
        self.steps = 0
        return self.state.copy()

    def step(self, action):
        x, x_dot, theta, theta_dot = self.state
        force = self.force_mag if action == 1 else -self.force_mag
        costheta = math.cos(theta)
        sintheta = math.sin(theta)
        temp = (force + self.polemass_length * theta_dot**2 * sintheta) / self.total_mass
        thetaacc = (self.gravity * sintheta - costheta * temp) / (
            self.length * (4.0 / 3.0 - self.masspole * costheta**2 / self.total_mass)
        )
        xacc = temp - self.polemass_length * thetaacc * costheta / self.total_mass
        x = x + self.tau * x_dot
        x_dot = x_dot + self.tau * xacc
        theta = theta + self.tau * theta_dot
        theta_dot = theta_dot + self.tau * thetaacc
        self.state = np.array([x, x_dot, theta, theta_dot], dtype=np.float32)
        self.steps += 1
        done = (
            x < -self.x_threshold
            or x > self.x_threshold
            or theta < -self.theta_threshold
            or theta > self.theta_threshold
            or self.steps >= self.max_steps
        )
        return self.state.copy(), 1.0, bool(done), {}


class SyntheticAtariEnv:
    """Synthetic Atari-shaped env: uint8 [4, 84, 84] observations.

    Observations come from a shared pre-generated frame bank (cheap per
    step); episode length ~ Geometric(1/mean_episode_len); reward sparse
    +-1. Deterministic per seed.
    """

    _bank = None

    def __init__(self, num_actions=18, mean_episode_len=1000, seed=None, bank_size=64):
        if SyntheticAtariEnv._bank is None:
            rng = np.random.RandomState(1234)
            SyntheticAtariEnv._bank = rng.randint(
                0, 256, size=(bank_size, 4, 84, 84), dtype=np.uint8
            )
        self.num_actions = num_actions
        self.p_done = 1.0 / mean_episode_len
        self.rng = np.random.RandomState(seed)
        self.t = 0

    def reset(self):
        self.t = int(self.rng.randint(0, len(self._bank)))
        return self._bank[self.t]

    def step(self, action):
        self.t = (self.t + 1 + int(action) % 3) % len(self._bank)
        obs = self._bank[self.t]
        reward = float(self.rng.randint(-1, 2)) if self.rng.rand() < 0.05 else 0.0
        done = bool(self.rng.rand() < self.p_done)
        return obs, reward, done, {}
