"""SMAC 3s5z QMIX workload (BASELINE.json config #5 analog; reference
dizoo/smac/config/smac_3s5z_qmix_config.py, win rate >= 0.999 on real SMAC).
Runs on the cooperative-matrix env with 8 agents / 14 actions matching the
3s5z agent count and action arity shape."""
from ding.utils import EasyDict

agent_num = 8
smac_3s5z_qmix_config = dict(
    exp_name='smac_3s5z_qmix_seed0',
    env=dict(
        agent_num=agent_num,
        action_dim=14,
        obs_dim=32,
        collector_env_num=8,
        evaluator_env_num=8,
        n_evaluator_episode=8,
        stop_value=0.999 * 25,  # per-step match fraction x episode length
    ),
    policy=dict(
        cuda=True,
        model=dict(
            agent_num=agent_num,
            obs_shape=32,
            global_obs_shape=agent_num * 14,
            action_shape=14,
            hidden_size_list=[64, 64],
            mixer=True,
        ),
        learn=dict(
            update_per_collect=20,
            batch_size=32,
            learning_rate=0.0005,
            clip_value=100,
            target_update_theta=0.008,
            discount_factor=0.99,
            double_q=False,
        ),
        collect=dict(n_sample=32, unroll_len=10, env_num=8),
        eval=dict(env_num=8, evaluator=dict(eval_freq=100, )),
        other=dict(
            eps=dict(type='exp', start=1, end=0.05, decay=50000),
            replay_buffer=dict(replay_buffer_size=5000, ),
        ),
    ),
)
main_config = EasyDict(smac_3s5z_qmix_config)
smac_3s5z_qmix_create_config = dict(
    env=dict(type='coop_matrix', import_names=['dizoo.multiagent.envs.coop_matrix_env']),
    env_manager=dict(type='base'),
    policy=dict(type='qmix'),
)
create_config = EasyDict(smac_3s5z_qmix_create_config)
