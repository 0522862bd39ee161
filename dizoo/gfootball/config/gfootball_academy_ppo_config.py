"""gfootball academy_empty_goal_close off-policy PPO (reference
dizoo/gfootball/entry/gfootball_ppo_config.py: simple115 obs, 19 actions)."""
from ding.utils import EasyDict

gfootball_academy_ppo_config = EasyDict(dict(
    exp_name='gfootball_academy_ppo_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=0.95,
    ),
    policy=dict(
        cuda=True,
        action_space='discrete',
        model=dict(obs_shape=115, action_shape=19, action_space='discrete',
                   encoder_hidden_size_list=[256, 128, 64]),
        learn=dict(update_per_collect=10, batch_size=320, learning_rate=3e-4, value_weight=0.5,
                   entropy_weight=0.01, clip_ratio=0.2, adv_norm=True),
        collect=dict(n_sample=3200, unroll_len=1, discount_factor=0.99, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=1000, )),
        other=dict(replay_buffer=dict(replay_buffer_size=10000)),
    ),
))
main_config = gfootball_academy_ppo_config
gfootball_academy_ppo_create_config = EasyDict(dict(
    env=dict(type='gfootball', import_names=['dizoo.gfootball.envs.gfootball_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ppo_offpolicy'),
))
create_config = gfootball_academy_ppo_create_config
