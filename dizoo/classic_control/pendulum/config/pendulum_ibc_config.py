"""Pendulum implicit behavioral cloning (EBM + Langevin MCMC; reference
dizoo/d4rl kitchen ibc configs adapted to the offline pendulum dataset)."""
from ding.utils import EasyDict

pendulum_ibc_config = EasyDict(dict(
    exp_name='pendulum_ibc_seed0',
    env=dict(
        collector_env_num=1,
        evaluator_env_num=5,
        n_evaluator_episode=5,
        stop_value=-250,
        act_scale=True,
    ),
    policy=dict(
        cuda=True,
        model=dict(obs_shape=3, action_shape=1, hidden_size=256, hidden_layer_num=3),
        learn=dict(batch_size=256, learning_rate=1e-4, update_per_collect=1),
        collect=dict(data_type='hdf5', data_path='./d4rl_data/pendulum-expert.npz', unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=1000, )),
    ),
))
main_config = pendulum_ibc_config
pendulum_ibc_create_config = EasyDict(dict(
    env=dict(type='pendulum', import_names=['dizoo.classic_control.pendulum.envs.pendulum_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ibc'),
))
create_config = pendulum_ibc_create_config
