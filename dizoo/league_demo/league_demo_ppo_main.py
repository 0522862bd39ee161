"""League-training loop on GameEnv: OneVsOneLeague hands out battle jobs
(PFSP over historical snapshots or self-play), the battle collector gathers
both streams, only the active player learns, results feed the payoff table
and snapshots freeze the learner into new historical opponents.

Parity: reference dizoo/league_demo/league_demo_ppo_main.py expressed with
this build's league + battle worker classes.
"""
import copy

import numpy as np
import torch

from ding.config import compile_config
from ding.envs import create_env_manager
from ding.league import create_league
from ding.policy import PPOPolicy
from ding.utils import EasyDict, deep_merge_dicts, set_pkg_seed
from ding.worker import BattleSampleSerialCollector
from dizoo.league_demo.game_env import GameEnv
from dizoo.league_demo.league_demo_ppo_config import league_demo_ppo_config
from dizoo.league_demo.selfplay_demo_ppo_main import _FlattenPPO


def main(cfg=None, seed: int = 0, max_train_iter: int = 40):
    cfg = EasyDict(copy.deepcopy(cfg if cfg is not None else league_demo_ppo_config))
    cfg = compile_config(cfg, seed=seed)
    set_pkg_seed(seed)
    league = create_league(cfg.league)
    pid = league.active_players_ids[0]

    pol_cfg = deep_merge_dicts(PPOPolicy.default_config(), cfg.policy)
    learner_policy = PPOPolicy(pol_cfg, enable_field=['learn', 'collect'])
    opponent_policy = PPOPolicy(pol_cfg, enable_field=['collect'])

    env_cfg = {'game_type': cfg.env.env_type, 'repeat_count': cfg.env.repeat_count}
    collector_env = create_env_manager(
        EasyDict({'type': 'base'}), [lambda: GameEnv(dict(env_cfg)) for _ in range(cfg.env.collector_env_num)]
    )
    collector = BattleSampleSerialCollector(
        EasyDict({'type': 'sample_1v1'}), env=collector_env,
        policy=[_FlattenPPO(learner_policy.collect_mode), _FlattenPPO(opponent_policy.collect_mode)],
        exp_name=cfg.exp_name
    )

    snapshots = {}  # player_id -> state_dict of the frozen learner

    for it in range(max_train_iter):
        job = league.get_job_info(pid)
        opp_id = job['player_id'][1]
        if opp_id in snapshots:
            opponent_policy.collect_mode.load_state_dict({'model': snapshots[opp_id]})
        else:  # self-play branch or untrained snapshot: mirror the learner
            opponent_policy.collect_mode.load_state_dict(
                {'model': learner_policy.learn_mode.state_dict()['model']}
            )
        data, info = collector.collect(n_sample=cfg.policy.collect.n_sample)
        learner_policy.learn_mode.forward(data[0])
        results = [i['result'] for i in info[0] if 'result' in i]
        league.finish_job({'launch_player': pid, 'player_id': job['player_id'], 'result': results or ['draws']})
        league.update_active_player({'player_id': pid, 'train_iter': it + 1})
        if league.judge_snapshot(pid):
            # freeze the current learner as a new historical opponent
            hist = [p for p in [p.player_id for p in league.historical_players] if p not in snapshots]
            state = copy.deepcopy(learner_policy.learn_mode.state_dict()['model'])
            for h in hist:
                snapshots[h] = state
    collector.close()
    payoff = league.payoff
    return learner_policy, league, payoff


if __name__ == '__main__':
    main()
