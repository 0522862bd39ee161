"""D4RL evaluation envs (hopper/halfcheetah/walker2d '-medium-v2' etc.).

The real D4RL/MuJoCo stack is unavailable offline; evaluation happens on the
native mujoco-lite dynamics (dizoo/mujoco/envs/mujoco_lite_env.py) with the
D4RL env-id naming preserved (reference dizoo/d4rl/envs/d4rl_env.py). The
matching offline datasets are synthesized by dizoo/d4rl/generate.py from
scripted behaviour policies of graded quality on the same dynamics, so
"medium" data really is mediocre relative to "expert" on the eval env.
"""
from ding.utils import ENV_REGISTRY
from dizoo.mujoco.envs.mujoco_lite_env import MujocoLiteEnv

_SHAPES = {
    'hopper': 'Hopper-v3',
    'halfcheetah': 'HalfCheetah-v3',
    'walker2d': 'Walker2d-v3',
}


@ENV_REGISTRY.register('d4rl')
class D4RLLiteEnv(MujocoLiteEnv):

    def __init__(self, cfg: dict = None) -> None:
        cfg = dict(cfg or {})
        env_id = cfg.get('env_id', 'hopper-medium-v2')
        family = env_id.split('-')[0].lower()
        cfg['env_id'] = _SHAPES.get(family, 'Hopper-v3')
        super().__init__(cfg)
        self._d4rl_id = env_id

    def __repr__(self) -> str:
        return f"D4RLLiteEnv({self._d4rl_id})"
