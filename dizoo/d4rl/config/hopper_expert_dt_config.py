"""D4RL hopper-expert Decision Transformer (reference
dizoo/d4rl/config/hopper_expert_dt_config.py; trajectory dataset synthesized
offline by dizoo/d4rl/generate.py)."""
from ding.utils import EasyDict

hopper_expert_dt_config = EasyDict(dict(
    exp_name='hopper_expert_dt_seed0',
    env=dict(
        env_id='hopper-expert-v2',
        collector_env_num=1,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=6000,
        use_act_scale=True,
    ),
    dataset=dict(
        data_dir_prefix='./d4rl_data/hopper-expert-v2.npz',
        context_len=20,
        rtg_scale=1000,
    ),
    policy=dict(
        cuda=True,
        obs_shape=11,
        action_shape=3,
        rtg_target=3600,
        max_eval_ep_len=1000,
        rtg_scale=1000,
        context_len=20,
        model=dict(
            state_dim=11,
            act_dim=3,
            n_blocks=3,
            h_dim=128,
            context_len=20,
            n_heads=1,
            drop_p=0.1,
            continuous=True,
        ),
        learn=dict(
            batch_size=64,
            learning_rate=1e-4,
            wt_decay=1e-4,
            warmup_steps=10000,
            clip_grad_norm_p=0.25,
        ),
        collect=dict(unroll_len=1, ),
        eval=dict(evaluator=dict(eval_freq=1000, )),
    ),
))
main_config = hopper_expert_dt_config
hopper_expert_dt_create_config = EasyDict(dict(
    env=dict(type='d4rl', import_names=['dizoo.d4rl.envs.d4rl_env']),
    env_manager=dict(type='base'),
    policy=dict(type='dt'),
))
create_config = hopper_expert_dt_create_config
