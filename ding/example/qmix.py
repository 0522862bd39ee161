"""Cooperative-matrix QMIX via the serial pipeline (SMAC-style obs dict)."""
from ding.entry import serial_pipeline
from ding.utils import EasyDict


def main(max_train_iter: int = 1000):
    main_config = EasyDict(dict(
        exp_name='exp/example_qmix',
        env=dict(collector_env_num=4, evaluator_env_num=4, n_evaluator_episode=4, stop_value=0.95,
                 agent_num=3, action_dim=4, obs_dim=8),
        policy=dict(
            cuda=False, priority=False,
            model=dict(agent_num=3, obs_shape=8, global_obs_shape=12, action_shape=4,
                       hidden_size_list=[64, 64, 64]),
            learn=dict(update_per_collect=4, batch_size=16, learning_rate=5e-4, target_update_theta=0.01,
                       discount_factor=0.99),
            collect=dict(n_sample=32, unroll_len=10, env_num=4),
            eval=dict(evaluator=dict(eval_freq=200), env_num=4),
            other=dict(eps=dict(type='exp', start=1.0, end=0.05, decay=50000),
                       replay_buffer=dict(replay_buffer_size=5000)),
        ),
    ))
    create_config = EasyDict(dict(
        env=dict(type='coop_matrix', import_names=['dizoo.multiagent.envs.coop_matrix_env']),
        env_manager=dict(type='base'),
        policy=dict(type='qmix'),
    ))
    return serial_pipeline((main_config, create_config), seed=0, max_train_iter=max_train_iter)


if __name__ == '__main__':
    main()
