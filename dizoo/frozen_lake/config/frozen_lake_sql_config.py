"""FrozenLake SQL (soft Q-learning) — tabular-friendly workload giving the
'sql' policy a runnable dizoo config."""
from ding.utils import EasyDict

frozen_lake_sql_config = EasyDict(dict(
    exp_name='frozen_lake_sql_seed0',
    env=dict(
        collector_env_num=8,
        evaluator_env_num=5,
        n_evaluator_episode=10,
        env_id='FrozenLake-v1',
        desc=None,
        map_name='4x4',
        is_slippery=False,
        stop_value=0.95,
    ),
    policy=dict(
        cuda=False,
        model=dict(obs_shape=16, action_shape=4, encoder_hidden_size_list=[128, 128, 64]),
        nstep=1,
        discount_factor=0.97,
        learn=dict(update_per_collect=5, batch_size=64, learning_rate=0.001, alpha=0.1),
        collect=dict(n_sample=10),
        eval=dict(evaluator=dict(eval_freq=40, )),
        other=dict(
            eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
            replay_buffer=dict(replay_buffer_size=20000, ),
        ),
    ),
))
main_config = frozen_lake_sql_config
frozen_lake_sql_create_config = EasyDict(dict(
    env=dict(type='frozen_lake', import_names=['dizoo.frozen_lake.envs.frozen_lake_env']),
    env_manager=dict(type='subprocess'),
    policy=dict(type='sql'),
))
create_config = frozen_lake_sql_create_config
