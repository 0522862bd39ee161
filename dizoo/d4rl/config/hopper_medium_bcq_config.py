"""D4RL bcq on hopper-medium-v2 (reference dizoo/d4rl/config/hopper_medium_bcq_config.py; dataset
synthesized offline by dizoo/d4rl/generate.py)."""
from ding.utils import EasyDict

hopper_medium_bcq_config = EasyDict(dict(
    exp_name='hopper_medium_bcq_config_seed0',
    env=dict(
        env_id='hopper-medium-v2',
        collector_env_num=1,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=6000,
        use_act_scale=True,
    ),
    policy=dict(
        cuda=True,
        model=dict(obs_shape=11, action_shape=3, action_space='regression', twin_critic=True, actor_head_hidden_size=512, critic_head_hidden_size=512),
        learn=dict(
            learning_rate_q=3e-4, learning_rate_policy=3e-4, learning_rate_vae=3e-4, lmbda=0.75,
            batch_size=256,
            update_per_collect=1,
        ),
        collect=dict(
            data_type='hdf5',
            data_path='./d4rl_data/hopper-medium-v2.npz',
            unroll_len=1,
            normalize_states=True,
        ),
        eval=dict(evaluator=dict(eval_freq=1000, )),
    ),
))
main_config = hopper_medium_bcq_config
hopper_medium_bcq_create_config = EasyDict(dict(
    env=dict(type='d4rl', import_names=['dizoo.d4rl.envs.d4rl_env']),
    env_manager=dict(type='base'),
    policy=dict(type='bcq'),
))
create_config = hopper_medium_bcq_create_config
