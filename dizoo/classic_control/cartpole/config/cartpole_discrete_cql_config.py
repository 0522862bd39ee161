"""CartPole discrete CQL on an offline dataset (reference
dizoo/cartpole cql configs; dataset synthesized by the smoke suite)."""
from ding.utils import EasyDict

cartpole_discrete_cql_config = EasyDict(dict(
    exp_name='cartpole_discrete_cql_seed0',
    env=dict(
        collector_env_num=1,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=195,
    ),
    policy=dict(
        cuda=True,
        nstep=1,
        model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[128, 128, 64],
                   num_quantiles=64),
        learn=dict(batch_size=64, learning_rate=1e-4, update_per_collect=1, min_q_weight=4.0),
        collect=dict(data_type='hdf5', data_path='./d4rl_data/cartpole-replay.npz', unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=1000, )),
    ),
))
main_config = cartpole_discrete_cql_config
cartpole_discrete_cql_create_config = EasyDict(dict(
    env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
    env_manager=dict(type='base'),
    policy=dict(type='discrete_cql'),
))
create_config = cartpole_discrete_cql_create_config
