"""D4RL halfcheetah-medium QGPO (contrastive energy-guided diffusion
policy; reference dizoo/d4rl/config/halfcheetah_medium_expert_qgpo_config)."""
from ding.utils import EasyDict

halfcheetah_medium_qgpo_config = EasyDict(dict(
    exp_name='halfcheetah_medium_qgpo_seed0',
    env=dict(
        env_id='halfcheetah-medium-v2',
        collector_env_num=1,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=6000,
        use_act_scale=True,
    ),
    policy=dict(
        cuda=True,
        model=dict(obs_shape=17, action_shape=6),
        learn=dict(batch_size=256, learning_rate=1e-4, update_per_collect=1),
        collect=dict(data_type='hdf5', data_path='./d4rl_data/halfcheetah-medium-v2.npz', unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=1000, )),
    ),
))
main_config = halfcheetah_medium_qgpo_config
halfcheetah_medium_qgpo_create_config = EasyDict(dict(
    env=dict(type='d4rl', import_names=['dizoo.d4rl.envs.d4rl_env']),
    env_manager=dict(type='base'),
    policy=dict(type='qgpo'),
))
create_config = halfcheetah_medium_qgpo_create_config
