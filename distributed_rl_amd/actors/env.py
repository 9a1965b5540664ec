"""Environments for the actor fleet.

The reference steps gym ALE Pong with manual preprocessing
(APE_X/Player.py:161-180,215-239: grayscale, NEAREST-resize to 84x84,
4-frame stack, frameskip 4, reward clip, life-loss pseudo-done). gym is not
installable in this image, so:

  * SyntheticEnv — Atari-shaped synthetic MDP used by tests and benchmarks
    (BASELINE.json: "synthetic 84x84x4 frames / random-init weights"). It has
    deterministic per-seed dynamics and a reward correlated with actions so
    learning sanity checks are possible.
  * AtariEnv — thin adapter over gym's ALE with the reference's exact
    preprocessing, used automatically when gym is importable.

Both expose: reset() -> state(4,84,84 u8); step(a) -> (state, reward, done, info).
"""

from __future__ import annotations

from typing import Tuple

import numpy as np

FRAME_SHAPE = (84, 84)
STACK = 4


class SyntheticEnv:
    """Cheap Atari-shaped MDP: observation is procedurally generated uint8
    noise whose mean encodes a hidden scalar state; reward = +1 when the
    action matches (hidden_state mod action_n), else small negative."""

    def __init__(self, action_n: int = 6, seed: int = 0,
                 episode_len: int = 512):
        self.action_n = action_n
        self.rng = np.random.default_rng(seed)
        self.episode_len = episode_len
        self._t = 0
        self._hidden = 0
        self._stack = np.zeros((STACK, *FRAME_SHAPE), dtype=np.uint8)

    def _frame(self) -> np.ndarray:
        # frame mean encodes (hidden mod action_n) exactly, so the optimal
        # policy is decodable from pixels (learning sanity checks depend on
        # this: tools/learning_sanity.py)
        base = 20 + (self._hidden % self.action_n) * 32
        f = self.rng.integers(0, 48, size=FRAME_SHAPE, dtype=np.uint8) + base
        return f.astype(np.uint8)

    def reset(self) -> np.ndarray:
        self._t = 0
        self._hidden = int(self.rng.integers(0, 1000))
        f = self._frame()
        for i in range(STACK):
            self._stack[i] = f
        return self._stack.copy()

    def step(self, action: int) -> Tuple[np.ndarray, float, bool, dict]:
        self._t += 1
        good = (self._hidden % self.action_n) == int(action)
        reward = 1.0 if good else -0.1
        self._hidden = (self._hidden * 1103515245 + 12345) % 1000
        self._stack[:-1] = self._stack[1:]
        self._stack[-1] = self._frame()
        done = self._t >= self.episode_len
        return self._stack.copy(), reward, done, {}

    @property
    def lives(self) -> int:
        return 1


class AtariEnv:
    """Reference-parity ALE wrapper (requires gym + atari; optional).

    Preprocessing parity with APE_X/Player.py:161-239: RGB->L grayscale,
    NEAREST resize to 84x84, frame stack 4, manual frameskip 4 (3 repeated
    steps + 1 observed), reward clip to [-1,1] (flag), life-loss pseudo-done.
    """

    def __init__(self, game: str = "PongNoFrameskip-v4", seed: int = 0,
                 reward_clip: bool = True, frame_skip: int = 4):
        import gym  # noqa — optional dependency

        self.sim = gym.make(game)
        try:
            self.sim.seed(seed)
        except Exception:
            pass
        self.reward_clip = reward_clip
        self.frame_skip = frame_skip
        self._stack = np.zeros((STACK, *FRAME_SHAPE), dtype=np.uint8)
        self._lives = None

    def _preprocess(self, obs: np.ndarray) -> np.ndarray:
        from PIL import Image

        img = Image.fromarray(obs).convert("L").resize(
            FRAME_SHAPE[::-1], Image.NEAREST
        )
        return np.asarray(img, dtype=np.uint8)

    def reset(self) -> np.ndarray:
        obs = self.sim.reset()
        if isinstance(obs, tuple):
            obs = obs[0]
        f = self._preprocess(obs)
        for i in range(STACK):
            self._stack[i] = f
        self._lives = None
        return self._stack.copy()

    def step(self, action: int):
        total_r = 0.0
        done = False
        info: dict = {}
        obs = None
        for _ in range(self.frame_skip):
            out = self.sim.step(action)
            if len(out) == 5:
                obs, r, term, trunc, info = out
                done = term or trunc
            else:
                obs, r, done, info = out
            total_r += float(r)
            if done:
                break
        if self.reward_clip:
            total_r = float(np.clip(total_r, -1.0, 1.0))
        # life-loss pseudo-done (APE_X/Player.py:227-239)
        lives = info.get("ale.lives", info.get("lives"))
        pseudo_done = done
        if lives is not None:
            if self._lives is not None and lives < self._lives:
                pseudo_done = True
            self._lives = lives
        self._stack[:-1] = self._stack[1:]
        self._stack[-1] = self._preprocess(obs)
        return self._stack.copy(), total_r, done, {"pseudo_done": pseudo_done, **info}


def make_env(kind: str = "auto", reward_clip: bool = True, **kw):
    """reward_clip carries cfg USE_REWARD_CLIP (cfg/ape_x.json:25) to the
    real-env path; the synthetic env has no unclipped rewards to clip."""
    if kind == "synthetic":
        return SyntheticEnv(**kw)
    if kind == "pong":
        from .pong import PongEnv

        kw.pop("game", None)
        return PongEnv(**kw)
    if kind == "atari":
        return AtariEnv(reward_clip=reward_clip, **kw)
    # auto: atari when gym importable, else synthetic
    try:
        import gym  # noqa: F401

        return AtariEnv(reward_clip=reward_clip, **kw)
    except Exception:
        kw.pop("game", None)
        return SyntheticEnv(**kw)
