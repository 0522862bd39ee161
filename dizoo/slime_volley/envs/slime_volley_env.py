"""Slime Volleyball implemented natively (slimevolleygym unavailable
offline). Interface parity with the reference
dizoo/slime_volley/envs/slime_volley_env.py: obs 12 (agent/ball/opponent
x,y,vx,vy from each side's own perspective), 6 discrete actions
({noop,left,right} x {ground,jump}), lives-based scoring (+-1 per rally,
5 lives, stop_value 5). ``agent_vs_bot`` plays a tracking heuristic;
``agent_vs_agent`` takes a list of two actions and returns per-side obs —
the self-play / league pipelines consume that mode.
"""
from typing import Any, List, Union

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY

G = -1.6           # gravity (units/s^2), court x in [0, 4], net at 2
DT = 1.0 / 25.0
SLIME_SPEED = 1.8
JUMP_V = 1.4
BALL_R = 0.12
SLIME_R = 0.35


@ENV_REGISTRY.register('slime_volley')
class SlimeVolleyEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self._vs_agent = self._cfg.get('agent_vs_agent', False)
        self._max_step = self._cfg.get('max_step', 1500)
        self._lives = self._cfg.get('lives', 5)
        self._observation_space = Box(-np.inf, np.inf, (12, ))
        self._action_space = Discrete(6)
        self._reward_space = Box(-1.0, 1.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def reset(self) -> Union[np.ndarray, List[np.ndarray]]:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        self._p1 = np.array([1.0, 0.0, 0.0, 0.0])  # x, y, vx, vy (left side)
        self._p2 = np.array([3.0, 0.0, 0.0, 0.0])
        self._serve(direction=1 if self._rng.rand() < 0.5 else -1)
        self._lives1 = self._lives2 = self._lives
        self._step_count = 0
        self._eval_episode_return = 0.0
        if self._vs_agent:
            return [self._obs(1), self._obs(2)]
        return self._obs(1)

    def _serve(self, direction: int) -> None:
        x = 1.0 if direction > 0 else 3.0
        self._ball = np.array([x, 1.5, 0.3 * direction, 0.0])
        self._rally_steps = 0

    def _obs(self, side: int) -> np.ndarray:
        """Perspective obs: own slime first, x mirrored for side 2."""
        me, other = (self._p1, self._p2) if side == 1 else (self._p2, self._p1)
        b = self._ball
        if side == 1:
            return np.array([*me, *b, *other], dtype=np.float32)
        # mirror x about the net (x -> 4 - x, vx -> -vx)
        def mir(v):
            return [4.0 - v[0], v[1], -v[2], v[3]]
        return np.array([*mir(me), *mir(b), *mir(other)], dtype=np.float32)

    def _apply(self, p: np.ndarray, action: int, side: int) -> None:
        move = {0: 0, 1: -1, 2: 1, 3: 0, 4: -1, 5: 1}[action]
        if side == 2:
            move = -move  # mirrored controls
        jump = action >= 3
        p[2] = move * SLIME_SPEED
        if jump and p[1] <= 0.0:
            p[3] = JUMP_V
        p[3] += G * DT
        p[0] = np.clip(p[0] + p[2] * DT, 0.0 + SLIME_R, 4.0 - SLIME_R)
        lo, hi = (SLIME_R, 2.0 - SLIME_R) if side == 1 else (2.0 + SLIME_R, 4.0 - SLIME_R)
        p[0] = np.clip(p[0], lo, hi)
        p[1] = max(p[1] + p[3] * DT, 0.0)
        if p[1] == 0.0:
            p[3] = max(p[3], 0.0)

    def _bot_action(self) -> int:
        """Track the ball; jump when it is close and descending."""
        b, me = self._ball, self._p2
        if b[0] < 2.0:
            target = 3.0
        else:
            target = np.clip(b[0] + b[2] * 0.2, 2.0 + SLIME_R, 4.0 - SLIME_R)
        # raw world-frame move (side-2 _apply mirrors, so pre-mirror here)
        move = 1 if target > me[0] + 0.05 else (2 if target < me[0] - 0.05 else 0)
        jump = abs(b[0] - me[0]) < 0.5 and b[1] < 1.0 and b[3] < 0
        return move + (3 if jump else 0)

    def step(self, action: Union[np.ndarray, List[np.ndarray], int]) -> BaseEnvTimestep:
        if self._vs_agent:
            a1, a2 = action[0], action[1]
        else:
            a1, a2 = action, self._bot_action()

        def as_int(a):
            if isinstance(a, np.ndarray):
                return int(a.reshape(-1)[0])
            return int(a)
        self._apply(self._p1, as_int(a1), 1)
        self._apply(self._p2, as_int(a2), 2)

        b = self._ball
        b[3] += G * DT
        b[0] += b[2] * DT
        b[1] += b[3] * DT
        # walls
        if b[0] < BALL_R or b[0] > 4.0 - BALL_R:
            b[2] = -b[2]
            b[0] = np.clip(b[0], BALL_R, 4.0 - BALL_R)
        # net (x=2, height 0.5)
        if abs(b[0] - 2.0) < BALL_R and b[1] < 0.5:
            b[2] = -b[2]
            b[0] = 2.0 + np.sign(b[2]) * (BALL_R + 1e-3)
        # slime collisions: elastic-ish bounce up and away
        for p in (self._p1, self._p2):
            dx, dy = b[0] - p[0], b[1] - (p[1] + 0.2)
            if dx * dx + dy * dy < (BALL_R + SLIME_R) ** 2 and b[3] < 0:
                norm = max(np.hypot(dx, dy), 1e-6)
                # horizontal kick includes slime motion + serve-breaking jitter
                b[2] = 1.6 * dx / norm + 0.5 * p[2] + self._rng.uniform(-0.2, 0.2)
                b[3] = max(1.6 * dy / norm, 0.8)
                self._rally_steps = 0

        reward = 0.0
        done = False
        # anti-stall: a rally that exceeds 300 steps without a slime touch
        # scores against the side holding the ball (keeps episodes bounded)
        self._rally_steps = getattr(self, '_rally_steps', 0) + 1
        if self._rally_steps > 300:
            b[1] = 0.0
        if b[1] <= BALL_R:  # ball grounded: point against that side
            if b[0] < 2.0:
                reward = -1.0
                self._lives1 -= 1
                self._serve(direction=1)
            else:
                reward = 1.0
                self._lives2 -= 1
                self._serve(direction=-1)
            if self._lives1 <= 0 or self._lives2 <= 0:
                done = True
        self._step_count += 1
        if self._step_count >= self._max_step:
            done = True
        self._eval_episode_return += reward
        info = {}
        if done:
            info['eval_episode_return'] = self._eval_episode_return
            info['final_eval_reward'] = self._eval_episode_return
        if self._vs_agent:
            obs = [self._obs(1), self._obs(2)]
            rew = np.array([reward, -reward], dtype=np.float32)
            if done:
                # per-side episode info for battle collectors
                info = {
                    'eval_episode_return': [self._eval_episode_return, -self._eval_episode_return],
                    'result': 'wins' if self._eval_episode_return > 0 else
                              ('losses' if self._eval_episode_return < 0 else 'draws'),
                }
            return BaseEnvTimestep(obs, rew, done, info)
        return BaseEnvTimestep(self._obs(1), np.array([reward], dtype=np.float32), done, info)

    def close(self) -> None:
        pass

    def random_action(self) -> np.ndarray:
        return np.array([self._action_space.sample()], dtype=np.int64)

    @property
    def observation_space(self):
        return self._observation_space

    @property
    def action_space(self):
        return self._action_space

    @property
    def reward_space(self):
        return self._reward_space

    def __repr__(self) -> str:
        return "SlimeVolleyEnv"
