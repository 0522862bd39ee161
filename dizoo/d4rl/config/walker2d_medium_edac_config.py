"""D4RL edac on walker2d-medium-v2 (reference dizoo/d4rl/config/walker2d_medium_edac_config.py; dataset
synthesized offline by dizoo/d4rl/generate.py)."""
from ding.utils import EasyDict

walker2d_medium_edac_config = EasyDict(dict(
    exp_name='walker2d_medium_edac_config_seed0',
    env=dict(
        env_id='walker2d-medium-v2',
        collector_env_num=1,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=6000,
        use_act_scale=True,
    ),
    policy=dict(
        cuda=True,
        model=dict(obs_shape=17, action_shape=6, ensemble_num=10, actor_head_hidden_size=256, critic_head_hidden_size=256),
        learn=dict(
            learning_rate_q=3e-4, learning_rate_policy=3e-4, eta=1.0,
            batch_size=256,
            update_per_collect=1,
        ),
        collect=dict(
            data_type='hdf5',
            data_path='./d4rl_data/walker2d-medium-v2.npz',
            unroll_len=1,
            normalize_states=True,
        ),
        eval=dict(evaluator=dict(eval_freq=1000, )),
    ),
))
main_config = walker2d_medium_edac_config
walker2d_medium_edac_create_config = EasyDict(dict(
    env=dict(type='d4rl', import_names=['dizoo.d4rl.envs.d4rl_env']),
    env_manager=dict(type='base'),
    policy=dict(type='edac'),
))
create_config = walker2d_medium_edac_create_config
