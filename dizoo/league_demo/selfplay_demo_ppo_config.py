"""Self-play PPO on the repeated zero-sum GameEnv (reference
dizoo/league_demo/selfplay_demo_ppo_config.py). Two PPO learners battle each
other; see selfplay_demo_ppo_main.py for the runnable loop."""
from ding.utils import EasyDict

selfplay_demo_ppo_config = EasyDict(dict(
    exp_name='exp/selfplay_demo_ppo',
    env=dict(
        env_type='zero_sum',
        repeat_count=4,
        collector_env_num=4,
        evaluator_env_num=4,
        n_evaluator_episode=8,
        stop_value=1.0,
    ),
    policy=dict(
        cuda=False,
        action_space='discrete',
        model=dict(
            obs_shape=4,  # flattened 2x2 payoff observation
            action_shape=2,
            action_space='discrete',
            encoder_hidden_size_list=[32, 32],
            critic_head_hidden_size=32,
            actor_head_hidden_size=32,
        ),
        learn=dict(
            epoch_per_collect=2,
            batch_size=32,
            learning_rate=1e-3,
            value_weight=0.5,
            entropy_weight=0.01,
            clip_ratio=0.2,
            adv_norm=True,
        ),
        collect=dict(n_sample=64, unroll_len=1, discount_factor=0.9, gae_lambda=0.95),
        eval=dict(evaluator=dict(eval_freq=50, )),
    ),
))
main_config = selfplay_demo_ppo_config
selfplay_demo_ppo_create_config = EasyDict(dict(
    env=dict(type='league_demo_game', import_names=['dizoo.league_demo.game_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ppo'),
))
create_config = selfplay_demo_ppo_create_config
