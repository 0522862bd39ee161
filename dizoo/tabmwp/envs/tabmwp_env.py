"""TabMWP-style prompt-selection env (reference dizoo/tabmwp/envs/
tabmwp_arithmetic_env.py). The real dataset/GPT grader are not available
offline; this env synthesizes arithmetic table-QA problems in K topic
families and rewards picking an in-context example of the SAME family —
the signal the reference's prompt-PG policies learn (choose relevant
shots). Obs: {'train_sample': str, 'candidate_samples': [str] * K};
action: candidate index; 1-step episodes, reward {0, 1}.
"""
from typing import Any

import numpy as np

from ding.envs import BaseEnv, BaseEnvTimestep
from ding.envs.common.spaces import Box, Discrete
from ding.utils import ENV_REGISTRY

TOPICS = [
    ("add", "what is {a} plus {b} in the table"),
    ("sub", "what is {a} minus {b} in the table"),
    ("mul", "what is {a} times {b} from the table row"),
    ("max", "which table entry is larger {a} or {b}"),
]


@ENV_REGISTRY.register('tabmwp')
class TabMWPLiteEnv(BaseEnv):

    def __init__(self, cfg: dict = None) -> None:
        self._cfg = cfg or {}
        self.cand_number = int(self._cfg.get('cand_number', 4))
        self._observation_space = Box(0, 1, (1, ))  # textual obs: nominal space
        self._action_space = Discrete(self.cand_number)
        self._reward_space = Box(0.0, 1.0, (1, ))
        self._rng = np.random.RandomState()
        self._seed = None
        self._dynamic_seed = True

    def seed(self, seed: int, dynamic_seed: bool = True) -> None:
        self._seed = seed
        self._dynamic_seed = dynamic_seed

    def _problem(self, topic_id: int) -> str:
        name, tpl = TOPICS[topic_id]
        return tpl.format(a=self._rng.randint(2, 99), b=self._rng.randint(2, 99)) + f" {name}"

    def reset(self) -> dict:
        if self._seed is not None:
            seed = self._seed + self._rng.randint(0, 100) if self._dynamic_seed else self._seed
            self._rng = np.random.RandomState(seed)
        self._topic = self._rng.randint(0, len(TOPICS))
        cands, self._correct = [], self._rng.randint(0, self.cand_number)
        for i in range(self.cand_number):
            t = self._topic if i == self._correct else \
                int(self._rng.choice([x for x in range(len(TOPICS)) if x != self._topic]))
            cands.append(self._problem(t))
        self._eval_episode_return = 0.0
        return {'train_sample': self._problem(self._topic), 'candidate_samples': cands}

    def step(self, action: Any) -> BaseEnvTimestep:
        if hasattr(action, 'reshape'):
            action = int(np.asarray(action).reshape(-1)[0])
        reward = 1.0 if int(action) == self._correct else 0.0
        self._eval_episode_return += reward
        obs = self.reset()  # next problem (episode is 1 step)
        return BaseEnvTimestep(
            obs, np.array([reward], dtype=np.float32), True,
            {'eval_episode_return': self._eval_episode_return}
        )

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
        return "TabMWPLiteEnv"
