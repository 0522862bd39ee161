"""CartPole PPO-PG (PPO policy loss without a critic; reference
cartpole_ppo_pg_config.py)."""
from ding.utils import EasyDict

cartpole_ppo_pg_config = EasyDict(dict(
    exp_name='cartpole_ppo_pg_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=195,
    ),
    policy=dict(
        cuda=False,
        action_space='discrete',
        model=dict(obs_shape=4, action_shape=2, action_space='discrete'),
        learn=dict(epoch_per_collect=2, batch_size=64, learning_rate=3e-4,
                   entropy_weight=0.001, clip_ratio=0.2),
        collect=dict(unroll_len=1, discount_factor=0.99, n_episode=8,
                     collector=dict(get_train_sample=True, type='episode')),
        eval=dict(evaluator=dict(eval_freq=100, )),
    ),
))
main_config = cartpole_ppo_pg_config
cartpole_ppo_pg_create_config = EasyDict(dict(
    env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ppo_pg'),
))
create_config = cartpole_ppo_pg_create_config
