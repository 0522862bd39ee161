"""dmc2gym cartpole-balance DreamerV3 (reference
dizoo/dmc2gym/config/cartpole_balance/cartpole_balance_dreamer_config.py;
run with ding.entry.serial_pipeline_dreamer)."""
from ding.utils import EasyDict

cartpole_balance_dreamer_config = EasyDict(dict(
    exp_name='dmc2gym_cartpole_balance_dreamer_seed0',
    env=dict(
        env_id='dmc2gym-cartpole-balance',
        domain_name='cartpole',
        task_name='balance',
        from_pixels=False,
        frame_skip=1,
        collector_env_num=8,
        evaluator_env_num=4,
        n_evaluator_episode=4,
        stop_value=900,
    ),
    policy=dict(
        cuda=True,
        random_collect_size=2500,
        imag_horizon=15,
        model=dict(
            action_shape=1,
            actor_dist='normal',
            dyn_stoch=32,
            dyn_deter=512,
            dyn_discrete=32,
            units=512,
            actor_layers=2,
            value_layers=2,
        ),
        learn=dict(batch_size=16, batch_length=64, learning_rate=3e-5),
        collect=dict(n_sample=200, unroll_len=1, action_size=1, collect_dyn_sample=True),
        eval=dict(evaluator=dict(eval_freq=1000, )),
        other=dict(replay_buffer=dict(type='sequence', replay_buffer_size=500000)),
    ),
    world_model=dict(
        type='dreamer',
        import_names=['ding.world_model.dreamer'],
        pretrain=100,
        train_freq=2,
        cuda=True,
        model=dict(
            state_size=5,
            obs_type='vector',
            action_size=1,
            action_type='continuous',
            encoder_hidden_size_list=[256, 256],
            dyn_stoch=32,
            dyn_deter=512,
            dyn_hidden=512,
            dyn_discrete=32,
            units=512,
            reward_layers=2,
            discount_layers=2,
            image_dec_layers=2,
            batch_size=16,
            batch_length=64,
        ),
    ),
))
main_config = cartpole_balance_dreamer_config
cartpole_balance_dreamer_create_config = EasyDict(dict(
    env=dict(type='dmc2gym', import_names=['dizoo.dmc2gym.envs.dmc2gym_env']),
    env_manager=dict(type='base'),
    policy=dict(type='dreamer'),
    world_model=dict(type='dreamer', import_names=['ding.world_model.dreamer']),
))
create_config = cartpole_balance_dreamer_create_config
