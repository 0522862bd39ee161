"""Default wrapper stacks for DingEnvWrapper.

Parity: reference ding/envs/env/default_wrapper.py (get_default_wrappers:8).
"""
import copy
from typing import List, Optional

from ding.utils import EasyDict

eval_episode_return_wrapper = EasyDict(type='eval_episode_return')


def get_default_wrappers(env_wrapper_name: str, env_id: Optional[str] = None,
                         caller: str = 'collector') -> List[dict]:
    """Named wrapper presets: 'atari_default' (noop/skip/life/warp/scale/
    clip/stack), 'mujoco_default', 'gym_hybrid_default', 'default'."""
    assert caller in ('collector', 'evaluator'), caller
    if env_wrapper_name == 'mujoco_default':
        return [copy.deepcopy(eval_episode_return_wrapper)]
    if env_wrapper_name == 'atari_default':
        wrapper_list = [
            EasyDict(type='noop_reset', kwargs=dict(noop_max=30)),
            EasyDict(type='max_and_skip', kwargs=dict(skip=4)),
            EasyDict(type='episodic_life'),
        ]
        if env_id is not None and any(k in env_id for k in ('Pong', 'Qbert', 'SpaceInvader', 'Montezuma')):
            wrapper_list.append(EasyDict(type='fire_reset'))
        wrapper_list.append(EasyDict(type='warp_frame'))
        wrapper_list.append(EasyDict(type='scaled_float_frame'))
        if caller == 'collector':
            wrapper_list.append(EasyDict(type='clip_reward'))
        wrapper_list.append(EasyDict(type='frame_stack', kwargs=dict(n_frames=4)))
        wrapper_list.append(copy.deepcopy(eval_episode_return_wrapper))
        return wrapper_list
    if env_wrapper_name == 'gym_hybrid_default':
        return [
            EasyDict(type='gym_hybrid_dict_action'),
            copy.deepcopy(eval_episode_return_wrapper),
        ]
    if env_wrapper_name == 'default':
        return [copy.deepcopy(eval_episode_return_wrapper)]
    raise NotImplementedError(f"not supported env_wrapper_name: {env_wrapper_name}")
