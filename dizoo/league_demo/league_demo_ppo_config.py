"""League PPO on GameEnv (reference dizoo/league_demo/league_demo_ppo_config.py):
one active PPO player trained against PFSP-sampled historical snapshots via
OneVsOneLeague. See league_demo_ppo_main.py."""
from ding.utils import EasyDict

from dizoo.league_demo.selfplay_demo_ppo_config import selfplay_demo_ppo_config

league_demo_ppo_config = EasyDict(dict(
    exp_name='exp/league_demo_ppo',
    env=dict(selfplay_demo_ppo_config.env),
    policy=dict(selfplay_demo_ppo_config.policy),
    league=dict(
        league_type='one_vs_one',
        player_category=['default'],
        path_policy='exp/league_demo_ppo/policy',
        active_players=dict(naive_sp_player=1),
        naive_sp_player=dict(
            one_phase_step=10,
            branch_probs=dict(pfsp=0.5, sp=0.5),
            strong_win_rate=0.7,
        ),
        use_pretrain=False,
        use_pretrain_init_historical=False,
        payoff=dict(type='battle', decay=0.99, min_win_rate_games=4),
        metric=dict(mu=0, sigma=25 / 3, beta=25 / 3 / 2, tau=0.0, draw_probability=0.02),
    ),
))
main_config = league_demo_ppo_config
league_demo_ppo_create_config = EasyDict(dict(
    env=dict(type='league_demo_game', import_names=['dizoo.league_demo.game_env']),
    env_manager=dict(type='base'),
    policy=dict(type='ppo'),
))
create_config = league_demo_ppo_create_config
