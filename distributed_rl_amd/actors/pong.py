"""From-scratch pixel Pong — the real-game validation environment.

The reference's only success criterion is ALE Pong reward -21 -> +21
(/root/reference/APE_X/Player.py:272-277, APE_X/Learner.py:226-231). ALE is
not installable in this image (no network, no ale-py wheel, no ROMs), so
this module re-implements the GAME — two paddles, ball physics, scripted
opponent, first to 21 — with the same observation/action contract the
reference's preprocessing produces (APE_X/Player.py:161-180,215-239):

  * observations: 4-frame stack of 84x84 uint8 grayscale, one observed
    frame per 4 simulation ticks (the reference's manual frameskip 4),
  * 6 actions with the ALE Pong meaning: NOOP/FIRE do nothing, RIGHT(2)
    and RIGHTFIRE(4) move up, LEFT(3) and LEFTFIRE(5) move down,
  * reward +1 / -1 per point, episode ends when either side reaches 21,
  * a uniformly random policy loses ~-21, a good policy reaches +21 (the
    scripted opponent is imperfect: capped speed + per-rally aim error).

Used by tools/pong_learning.py to demonstrate the -21 -> positive learning
curve the reference treats as its integration test.
"""

from __future__ import annotations

from typing import Tuple

import numpy as np

H = W = 84
STACK = 4
TICKS_PER_STEP = 4  # frameskip
PADDLE_H = 7.0
PADDLE_W = 2
BALL = 2
AGENT_X = 78  # left edge of the agent paddle (right side)
OPP_X = 4  # left edge of the opponent paddle (left side)
PADDLE_SPEED = 1.2  # px per tick (agent)
OPP_SPEED = 0.72  # px per tick (slower than the agent -> beatable)
BALL_VX = 1.1  # serve speed, px per tick
BALL_VX_MAX = 1.9
WIN_SCORE = 21
BG, FG = 60, 236  # grayscale levels (net drawn dimmer than sprites)

# ALE Pong action semantics: 0 NOOP, 1 FIRE, 2 RIGHT(up), 3 LEFT(down),
# 4 RIGHTFIRE(up), 5 LEFTFIRE(down)
_ACTION_DY = {0: 0.0, 1: 0.0, 2: -1.0, 3: 1.0, 4: -1.0, 5: 1.0}


class PongEnv:
    def __init__(self, action_n: int = 6, seed: int = 0,
                 max_steps: int = 3000):
        assert action_n == 6, "Pong uses the 6-action ALE set"
        self.action_n = action_n
        self.rng = np.random.default_rng(seed)
        self.max_steps = max_steps
        self._stack = np.zeros((STACK, H, W), dtype=np.uint8)

    # -- dynamics ----------------------------------------------------------
    def _serve(self, towards: int):
        """towards: +1 serve to the agent (right), -1 to the opponent."""
        self.ball_x = W / 2.0
        self.ball_y = float(self.rng.uniform(20, H - 20))
        self.vx = BALL_VX * towards
        self.vy = float(self.rng.uniform(-1.0, 1.0))
        # per-rally opponent aim error (what makes it beatable)
        self.opp_err = float(self.rng.normal(0.0, 4.0))

    def reset(self) -> np.ndarray:
        self.agent_y = self.opp_y = H / 2.0  # paddle centers
        self.score_agent = self.score_opp = 0
        self.steps = 0
        self._agent_vy = 0.0
        self._serve(towards=1 if self.rng.random() < 0.5 else -1)
        f = self._render()
        for i in range(STACK):
            self._stack[i] = f
        return self._stack.copy()

    def _tick(self, dy: float) -> float:
        """One simulation tick; returns the point reward (0 if rally goes on)."""
        half = PADDLE_H / 2.0
        self._agent_vy = dy * PADDLE_SPEED
        self.agent_y = float(np.clip(self.agent_y + self._agent_vy,
                                     half, H - half))
        # opponent: track the ball (with aim error) while it approaches,
        # drift back to center while it recedes
        target = (self.ball_y + self.opp_err) if self.vx < 0 else H / 2.0
        delta = target - self.opp_y
        self.opp_y = float(np.clip(
            self.opp_y + np.clip(delta, -OPP_SPEED, OPP_SPEED), half, H - half
        ))

        self.ball_x += self.vx
        self.ball_y += self.vy
        # wall bounce
        if self.ball_y < 0:
            self.ball_y = -self.ball_y
            self.vy = -self.vy
        elif self.ball_y > H - BALL:
            self.ball_y = 2 * (H - BALL) - self.ball_y
            self.vy = -self.vy

        # paddle collisions (checked at the crossing tick)
        if (self.vx > 0 and AGENT_X <= self.ball_x + BALL
                and self.ball_x < AGENT_X + PADDLE_W):
            off = (self.ball_y + BALL / 2.0) - self.agent_y
            if abs(off) <= half + BALL / 2.0:
                self.ball_x = float(AGENT_X - BALL)
                self.vx = -min(abs(self.vx) * 1.04, BALL_VX_MAX)
                self.vy = 1.6 * off / half + 0.35 * self._agent_vy \
                    + float(self.rng.uniform(-0.08, 0.08))
        elif (self.vx < 0 and self.ball_x <= OPP_X + PADDLE_W
                and self.ball_x + BALL > OPP_X):
            off = (self.ball_y + BALL / 2.0) - self.opp_y
            if abs(off) <= half + BALL / 2.0:
                self.ball_x = float(OPP_X + PADDLE_W)
                self.vx = min(abs(self.vx) * 1.04, BALL_VX_MAX)
                self.vy = 1.6 * off / half + float(self.rng.uniform(-0.08, 0.08))

        # scoring
        if self.ball_x < 0:
            self.score_agent += 1
            self._serve(towards=-1)
            return 1.0
        if self.ball_x > W - BALL:
            self.score_opp += 1
            self._serve(towards=1)
            return -1.0
        return 0.0

    def _render(self) -> np.ndarray:
        f = np.zeros((H, W), dtype=np.uint8)
        f[:] = 0
        f[:, W // 2 : W // 2 + 1] = BG  # net
        half = int(PADDLE_H // 2)
        ay, oy = int(self.agent_y), int(self.opp_y)
        f[max(0, ay - half) : min(H, ay + half + 1),
          AGENT_X : AGENT_X + PADDLE_W] = FG
        f[max(0, oy - half) : min(H, oy + half + 1),
          OPP_X : OPP_X + PADDLE_W] = FG
        bx, by = int(self.ball_x), int(self.ball_y)
        f[max(0, by) : min(H, by + BALL), max(0, bx) : min(W, bx + BALL)] = FG
        return f

    def step(self, action: int) -> Tuple[np.ndarray, float, bool, dict]:
        dy = _ACTION_DY[int(action)]
        reward = 0.0
        for _ in range(TICKS_PER_STEP):
            reward += self._tick(dy)
        self.steps += 1
        self._stack[:-1] = self._stack[1:]
        self._stack[-1] = self._render()
        done = (self.score_agent >= WIN_SCORE or self.score_opp >= WIN_SCORE
                or self.steps >= self.max_steps)
        info = {"score": (self.score_agent, self.score_opp)}
        return self._stack.copy(), reward, done, info

    @property
    def lives(self) -> int:
        return 1
