"""Imitation-by-logit-regression (IL) on the league demo game (reference
dizoo/gfootball IL usage generalised: any expert data with obs+logit)."""
from ding.utils import EasyDict

league_demo_il_config = EasyDict(dict(
    exp_name='league_demo_il_seed0',
    env=dict(
        collector_env_num=2,
        evaluator_env_num=2,
        n_evaluator_episode=4,
        stop_value=0.9,
    ),
    policy=dict(
        cuda=False,
        model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[64, 64]),
        learn=dict(update_per_collect=10, batch_size=64, learning_rate=2e-4),
        collect=dict(unroll_len=1, discount_factor=0.99, n_sample=64),
        eval=dict(evaluator=dict(eval_freq=200, )),
        other=dict(replay_buffer=dict(replay_buffer_size=100000)),
    ),
))
main_config = league_demo_il_config
league_demo_il_create_config = EasyDict(dict(
    env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
    env_manager=dict(type='base'),
    policy=dict(type='IL'),
))
create_config = league_demo_il_create_config
