"""League manager: player lifecycle, job dispatch, payoff updates, snapshots.

Parity: reference ding/league/base_league.py (BaseLeague:17) and
one_vs_one_league.py.
"""
import copy
import os
from typing import Any, Dict, List

from ding.utils import LEAGUE_REGISTRY, EasyDict, deep_merge_dicts, LockContext, LockContextType, import_module
from .player import ActivePlayer, HistoricalPlayer, create_player, MainPlayer
from .shared_payoff import create_payoff


@LEAGUE_REGISTRY.register('base')
class BaseLeague:

    config = dict(
        league_type='base',
        import_names=[],
        player_category=['default'],
        active_players=dict(main_player=1),
        main_player=dict(
            one_phase_step=2e5,
            branch_probs=dict(pfsp=0.5, sp=0.5),
            strong_win_rate=0.7,
        ),
        payoff=dict(type='battle', decay=0.99, min_win_rate_games=8),
        metric=dict(mu=0, sigma=25 / 3),
        path_policy='league_policy',
    )

    @classmethod
    def default_config(cls) -> EasyDict:
        return EasyDict(copy.deepcopy(cls.config))

    def __init__(self, cfg: EasyDict):
        self.cfg = deep_merge_dicts(self.default_config(), cfg or EasyDict({}))
        # active_players is a choice, not a merge target: user config replaces it
        if cfg and 'active_players' in cfg:
            self.cfg.active_players = EasyDict(cfg.active_players)
        self.path_policy = self.cfg.path_policy
        os.makedirs(self.path_policy, exist_ok=True)
        self.active_players: List[ActivePlayer] = []
        self.historical_players: List[HistoricalPlayer] = []
        self.payoff = create_payoff(self.cfg.payoff)
        self._active_players_lock = LockContext(LockContextType.THREAD_LOCK)
        self._init_players()

    def _init_players(self) -> None:
        for cat in self.cfg.player_category:
            for player_type, num in self.cfg.active_players.items():
                for i in range(num):
                    name = f'{player_type}_{cat}_{i}'
                    ckpt_path = os.path.join(self.path_policy, name + '_ckpt.pth')
                    player_cfg = self.cfg.get(player_type, EasyDict({}))
                    player = create_player(
                        self.cfg, player_type, player_cfg, cat, self.payoff, ckpt_path, name, 0
                    )
                    self.active_players.append(player)
                    self.payoff.add_player(player)

    @property
    def active_players_ids(self) -> List[str]:
        return [p.player_id for p in self.active_players]

    @property
    def active_players_ckpts(self) -> List[str]:
        return [p.checkpoint_path for p in self.active_players]

    # ------------------------------------------------------------- jobs
    def get_job_info(self, player_id: str, eval_flag: bool = False) -> dict:
        player = self.get_player_by_id(player_id)
        job = player.get_job(eval_flag)
        opponent = job['opponent']
        if eval_flag:
            return {
                'agent_num': 1,
                'launch_player': player_id,
                'player_id': [player_id],
                'checkpoint_path': [player.checkpoint_path],
                'player_active_flag': [isinstance(player, ActivePlayer)],
                'eval_opponent': opponent,
            }
        return {
            'agent_num': 2,
            'launch_player': player_id,
            'player_id': [player_id, opponent.player_id],
            'checkpoint_path': [player.checkpoint_path, opponent.checkpoint_path],
            'player_active_flag': [isinstance(p, ActivePlayer) for p in [player, opponent]],
        }

    def judge_snapshot(self, player_id: str, force: bool = False) -> bool:
        player = self.get_player_by_id(player_id)
        if not isinstance(player, ActivePlayer):
            return False
        if force or player.is_trained_enough():
            hp = player.snapshot()
            self.historical_players.append(hp)
            self.payoff.add_player(hp)
            self._save_checkpoint(player.checkpoint_path, hp.checkpoint_path)
            # mutate (exploiters may reset)
            reset_path = player.mutate({'pretrain_checkpoint_path': player.checkpoint_path})
            if reset_path is not None:
                self.load_checkpoint(player_id, reset_path)
            return True
        return False

    def update_active_player(self, player_info: dict) -> None:
        player = self.get_player_by_id(player_info['player_id'])
        if isinstance(player, ActivePlayer):
            player.total_agent_step = player_info['train_iter']

    def finish_job(self, job_info: dict) -> None:
        self.payoff.update(job_info)

    def get_player_by_id(self, player_id: str):
        for p in self.active_players + self.historical_players:
            if p.player_id == player_id:
                return p
        raise KeyError(player_id)

    # ---------------------------------------------------------- ckpt glue
    def _save_checkpoint(self, src: str, dst: str) -> None:
        if os.path.exists(src):
            import shutil
            shutil.copy(src, dst)

    def load_checkpoint(self, player_id: str, path: str) -> None:
        pass  # learner-side hook

    def save_checkpoint(self, player_id: str, path: str) -> None:
        pass


@LEAGUE_REGISTRY.register('one_vs_one')
class OneVsOneLeague(BaseLeague):

    config = dict(
        league_type='one_vs_one',
        player_category=['default'],
        active_players=dict(main_player=1),
        main_player=dict(
            one_phase_step=2e5,
            branch_probs=dict(pfsp=0.2, sp=0.8),
            strong_win_rate=0.7,
        ),
        use_pretrain=False,
        use_pretrain_init_historical=False,
        payoff=dict(type='battle', decay=0.99, min_win_rate_games=8),
        metric=dict(mu=0, sigma=25 / 3),
        path_policy='league_policy',
    )


def create_league(cfg: EasyDict, *args) -> BaseLeague:
    cfg = EasyDict(cfg)
    import_module(cfg.get('import_names', []))
    return LEAGUE_REGISTRY.build(cfg.league_type, cfg=cfg, *args)
