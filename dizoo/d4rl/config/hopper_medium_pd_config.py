"""D4RL hopper-medium Plan Diffuser (reference
dizoo/d4rl/config/hopper_medium_pd_config.py; trajectory batches come from
the synthesized dataset, see tests/test_diffusion.py for the batch schema)."""
from ding.utils import EasyDict

HORIZON = 32
hopper_medium_pd_config = EasyDict(dict(
    exp_name='hopper_medium_pd_seed0',
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
        model=dict(
            diffuser_model='GaussianDiffusion',
            diffuser_model_cfg=dict(
                model='DiffusionUNet1d',
                model_cfg=dict(transition_dim=11 + 3, dim=32, dim_mults=[1, 2, 4]),
                horizon=HORIZON, obs_dim=11, action_dim=3, n_timesteps=20, clip_denoised=True,
            ),
            value_model='ValueDiffusion',
            value_model_cfg=dict(
                model='TemporalValue',
                model_cfg=dict(horizon=HORIZON, transition_dim=11 + 3, dim=32, dim_mults=[1, 2, 4]),
                horizon=HORIZON, obs_dim=11, action_dim=3, n_timesteps=20,
            ),
            scale=0.1, t_stopgrad=2, n_guide_steps=2,
        ),
        learn=dict(
            batch_size=64, learning_rate=2e-4, gradient_accumulate_every=2, plan_batch_size=64,
            update_target_freq=10, step_start_update_target=200, target_weight=0.995, value_step=200,
            include_returns=True, discount_factor=0.99, ignore_done=False,
        ),
        collect=dict(unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=1000, )),
        other=dict(replay_buffer=dict(replay_buffer_size=10000)),
    ),
))
main_config = hopper_medium_pd_config
hopper_medium_pd_create_config = EasyDict(dict(
    env=dict(type='d4rl', import_names=['dizoo.d4rl.envs.d4rl_env']),
    env_manager=dict(type='base'),
    policy=dict(type='pd'),
))
create_config = hopper_medium_pd_create_config
