"""Standard Atari preprocessing as a composable wrapper stack.

The reference wires ALE through gym wrappers in examples/atari/environment.py
(NoopReset, MaxAndSkip, EpisodicLife, FireReset, 84x84 grayscale, frame
stack, reward clipping) and feeds the result to EnvPool. Neither gym nor
ALE ROMs exist in this build environment, so `AtariPreprocessing` is
duck-typed: it wraps ANY object with `reset()/step(action)` returning
HxWx3 uint8 RGB frames (old- or new-style gym API) and produces exactly
the [4, 84, 84] uint8 observation the IMPALA benchmark consumes — the
same tensor `SyntheticAtariEnv` fabricates. Plug a real
`gym.make("...NoFrameskip-v4")` in here when ROMs are available; nothing
else in the framework changes.

All image work is numpy (CPU, inside EnvPool worker processes); the GPU
side of preprocessing (uint8 -> bf16 NHWC scale) stays in the fused HIP
kernels (hip/kernels.hip frames_u8_to_bf16_nhwc / conv1_u8_nhwc).
"""
import numpy as np


def _step(env, action):
    """Normalize old (obs, r, done, info) / new (obs, r, term, trunc, info)."""
    out = env.step(action)
    if len(out) == 5:
        obs, r, term, trunc, info = out
        return obs, r, bool(term) or bool(trunc), info
    obs, r, done, info = out
    return obs, r, bool(done), info


def _reset(env):
    out = env.reset()
    if isinstance(out, tuple) and len(out) == 2 and isinstance(out[1], dict):
        return out[0]
    return out


def rgb_to_gray(frame):
    """ITU-R 601-2 luma, uint8 in/out (matches cv2.cvtColor RGB2GRAY)."""
    f = np.asarray(frame)
    if f.ndim == 2:
        return f.astype(np.uint8, copy=False)
    w = np.array([0.299, 0.587, 0.114], dtype=np.float32)
    return (f.astype(np.float32) @ w).astype(np.uint8)


def resize_area(img, h, w):
    """INTER_AREA-style box resize for downscaling, uint8 in/out.

    210x160 -> 84x84 is non-integer, so we box-average on a fractional
    grid: each output pixel averages the input region it covers, with
    edge pixels weighted by coverage. Pure numpy (runs in env workers).
    """
    img = np.asarray(img, dtype=np.float32)
    ih, iw = img.shape[:2]

    def axis_weights(n_in, n_out):
        # sparse row-matrix W [n_out, n_in]: W @ x averages boxes
        scale = n_in / n_out
        W = np.zeros((n_out, n_in), dtype=np.float32)
        for o in range(n_out):
            a, b = o * scale, (o + 1) * scale
            lo, hi = int(np.floor(a)), int(np.ceil(b))
            for i in range(lo, min(hi, n_in)):
                cover = min(b, i + 1) - max(a, i)
                if cover > 0:
                    W[o, i] = cover
            W[o] /= W[o].sum()
        return W

    Wh = axis_weights(ih, h)
    Ww = axis_weights(iw, w)
    out = Wh @ img @ Ww.T
    return np.clip(out + 0.5, 0, 255).astype(np.uint8)


class AtariPreprocessing:
    """NoopReset + MaxAndSkip + EpisodicLife + FireReset + gray/84x84 +
    4-frame stack + reward clip, over any RGB-frame env.

    Observation: uint8 [4, 84, 84] (newest frame last). Exposes
    `action_space`/`observation_space`-free duck API for EnvPool.
    """

    def __init__(
        self,
        env,
        frame_skip=4,
        frame_stack=4,
        size=84,
        noop_max=30,
        episodic_life=True,
        clip_rewards=True,
        fire_reset=True,
        rng=None,
    ):
        self.env = env
        self.frame_skip = frame_skip
        self.frame_stack = frame_stack
        self.size = size
        self.noop_max = noop_max
        self.episodic_life = episodic_life
        self.clip_rewards = clip_rewards
        self.rng = rng if rng is not None else np.random.RandomState()
        self._stack = np.zeros((frame_stack, size, size), dtype=np.uint8)
        self._skip_buf = None  # last two raw frames for flicker max
        self._lives = 0
        self._real_done = True
        meanings = getattr(env, "get_action_meanings", lambda: [])()
        self._fire = 1 if (fire_reset and "FIRE" in meanings) else None
        # cached resize weights (rebuilt if the frame size changes)
        self._w_cache = {}

    # -------------------------------------------------------------- frames

    def _observe(self, raw):
        if self._skip_buf is not None:
            raw = np.maximum(raw, self._skip_buf)  # ALE flicker removal
        gray = rgb_to_gray(raw)
        small = resize_area(gray, self.size, self.size)
        self._stack = np.roll(self._stack, -1, axis=0)
        self._stack[-1] = small
        return self._stack.copy()

    def _lives_of(self, info):
        if isinstance(info, dict) and "lives" in info:
            return info["lives"]
        ale = getattr(self.env, "ale", None)
        if ale is not None:
            return ale.lives()
        return 0

    # ----------------------------------------------------------------- api

    def reset(self):
        if self._real_done or not self.episodic_life:
            raw = _reset(self.env)
            for _ in range(int(self.rng.randint(0, self.noop_max + 1)) if self.noop_max else 0):
                raw, _, done, _ = _step(self.env, 0)
                if done:
                    raw = _reset(self.env)
            if self._fire is not None:
                raw, _, done, _ = _step(self.env, self._fire)
                if done:
                    raw = _reset(self.env)
            self._stack[:] = 0
        else:
            # episodic-life soft reset: keep playing, keep the stack
            raw, _, done, info = _step(self.env, 0)
            if done:
                self._real_done = True
                return self.reset()
        self._skip_buf = None
        self._lives = self._lives_of({})
        self._real_done = False
        return self._observe(np.asarray(raw))

    def step(self, action):
        total_r = 0.0
        done = False
        info = {}
        raw = None
        for i in range(self.frame_skip):
            prev = raw
            raw, r, done, info = _step(self.env, action)
            total_r += float(r)
            if i == self.frame_skip - 2:
                self._skip_buf = np.asarray(raw).copy()
            if done:
                break
        self._real_done = done
        if self.episodic_life:
            lives = self._lives_of(info)
            if 0 < lives < self._lives:
                done = True
            self._lives = lives
        obs = self._observe(np.asarray(raw))
        self._skip_buf = None
        if self.clip_rewards:
            total_r = float(np.sign(total_r))
        return obs, total_r, done, info

    def close(self):
        close = getattr(self.env, "close", None)
        if close is not None:
            close()
