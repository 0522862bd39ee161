"""Behavioral matrix for the classic env wrappers (reference
ding/envs/env_wrappers/tests): noop reset, max-and-skip, episodic life,
fire reset, frame stack, clip reward, time limit, delayed reward,
eval-episode-return accounting, running obs/reward norm.
"""
import numpy as np
import pytest

from ding.envs.env_wrappers.env_wrappers import (
    ClipRewardWrapper, DelayRewardWrapper, EpisodicLifeWrapper, EvalEpisodeReturnWrapper, FireResetWrapper,
    FrameStackWrapper, MaxAndSkipWrapper, NoopResetWrapper, ObsNormWrapper, RewardNormWrapper, TimeLimitWrapper,
)


class ScriptEnv:
    """Deterministic scripted env: obs counts steps; configurable lives/fire."""

    def __init__(self, lives=1, obs_shape=(2, ), needs_fire=False):
        self.t = 0
        self.resets = 0
        self.actions = []
        self.lives_total = lives
        self.lives_left = lives
        self.needs_fire = needs_fire
        self.obs_shape = obs_shape

        class _AS:
            n = 4

            def sample(self):
                return 0

        self.action_space = _AS()
        self.unwrapped = self
        self.observation_space = None

    def get_action_meanings(self):
        return ['NOOP', 'FIRE', 'RIGHT', 'LEFT'] if self.needs_fire else ['NOOP', 'RIGHT', 'LEFT', 'UP']

    @property
    def ale(self):
        outer = self

        class _Ale:

            def lives(self):
                return outer.lives_left

        return _Ale()

    def reset(self, **kw):
        self.resets += 1
        self.t = 0
        self.lives_left = self.lives_total
        return np.full(self.obs_shape, float(self.t), dtype=np.float32)

    def step(self, action):
        self.actions.append(int(action))
        self.t += 1
        if self.t % 5 == 0 and self.lives_left > 0:
            self.lives_left -= 1
        done = self.lives_left == 0
        obs = np.full(self.obs_shape, float(self.t), dtype=np.float32)
        return obs, float(self.t), done, {'lives': self.lives_left}


def test_noop_reset_executes_noops():
    env = ScriptEnv(lives=100)
    w = NoopResetWrapper(env, noop_max=7)
    w.reset()
    assert 1 <= env.t <= 7, "reset must advance 1..noop_max noop steps"
    assert all(a == 0 for a in env.actions)


def test_max_and_skip_repeats_action_and_sums_reward():
    env = ScriptEnv(lives=100)
    w = MaxAndSkipWrapper(env, skip=4)
    w.reset()
    obs, r, d, i = w.step(3)
    assert env.t == 4, "skip=4 must step the inner env 4 times"
    assert r == 1 + 2 + 3 + 4
    assert env.actions == [3, 3, 3, 3]
    assert float(obs.reshape(-1)[0]) == 4.0  # max of last two frames


def test_episodic_life_splits_episodes():
    env = ScriptEnv(lives=2)
    w = EpisodicLifeWrapper(env)
    w.reset()
    done_at = []
    for t in range(1, 11):
        _, _, done, _ = w.step(0)
        if done:
            done_at.append(env.t)
            w.reset()
    # inner lives drop at t=5 and t=10: wrapper reports done both times but
    # only the true game-over (lives==0) resets the inner env
    assert done_at[0] == 5
    assert env.resets == 2  # initial + the real game over


def test_fire_reset_presses_fire():
    env = ScriptEnv(lives=100, needs_fire=True)
    w = FireResetWrapper(env)
    w.reset()
    assert 1 in env.actions[:2], "FIRE must be pressed on reset"


def test_frame_stack_and_clip_reward():
    env = ScriptEnv(lives=100, obs_shape=(2, 2))
    w = ClipRewardWrapper(FrameStackWrapper(env, n_frames=4))
    obs = w.reset()
    assert np.asarray(obs).shape == (4, 2, 2)
    obs, r, d, i = w.step(0)
    assert r == 1.0  # clipped sign(1)
    assert float(np.asarray(obs)[-1, 0, 0]) == 1.0
    assert float(np.asarray(obs)[0, 0, 0]) == 0.0


def test_time_limit_truncates():
    env = ScriptEnv(lives=100)
    w = TimeLimitWrapper(env, max_limit=3)
    w.reset()
    dones = [w.step(0)[2] for _ in range(3)]
    assert dones == [False, False, True]


def test_delay_reward_accumulates():
    env = ScriptEnv(lives=100)
    w = DelayRewardWrapper(env, delay_reward_step=3)
    w.reset()
    rs = [w.step(0)[1] for _ in range(6)]
    assert rs[0] == 0 and rs[1] == 0 and rs[2] == 1 + 2 + 3
    assert rs[3] == 0 and rs[4] == 0 and rs[5] == 4 + 5 + 6


def test_eval_episode_return_accounting():
    env = ScriptEnv(lives=1)  # done at t=5
    w = EvalEpisodeReturnWrapper(env)
    w.reset()
    for _ in range(4):
        _, _, done, info = w.step(0)
        assert not done and 'eval_episode_return' not in info
    _, _, done, info = w.step(0)
    assert done and info['eval_episode_return'] == 1 + 2 + 3 + 4 + 5


def test_obs_and_reward_norm_track_statistics():
    env = ScriptEnv(lives=100)
    wo = ObsNormWrapper(env)
    wo.reset()
    outs = [wo.step(0)[0] for _ in range(50)]
    assert np.abs(outs[-1]).max() <= 10.0, "normalized obs must be bounded"
    env2 = ScriptEnv(lives=100)
    wr = RewardNormWrapper(env2, reward_discount=0.99)
    wr.reset()
    rs = [float(wr.step(0)[1]) for _ in range(50)]
    assert abs(rs[-1]) < abs(50.0), "reward norm must rescale the raw reward"
