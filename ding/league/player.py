"""League players.

Parity: reference ding/league/player.py (Player:10, ActivePlayer:89,
HistoricalPlayer:114), starcraft_player.py (MainPlayer/MainExploiter/
LeagueExploiter branch behavior).
"""
import uuid
from collections import namedtuple
from typing import Any, Callable, List, Optional

import numpy as np

from ding.utils import PLAYER_REGISTRY, EasyDict
from .algorithm import pfsp


class Player:

    _name = "BasePlayer"

    def __init__(self, cfg: EasyDict, category: str, init_payoff, checkpoint_path: str, player_id: str,
                 total_agent_step: int):
        self._cfg = cfg or EasyDict({})
        self._category = category
        self._payoff = init_payoff
        self._checkpoint_path = checkpoint_path
        assert isinstance(player_id, str)
        self._player_id = player_id
        self._total_agent_step = total_agent_step

    @property
    def category(self) -> str:
        return self._category

    @property
    def payoff(self):
        return self._payoff

    @property
    def checkpoint_path(self) -> str:
        return self._checkpoint_path

    @property
    def player_id(self) -> str:
        return self._player_id

    @property
    def total_agent_step(self) -> int:
        return self._total_agent_step

    @total_agent_step.setter
    def total_agent_step(self, step: int):
        self._total_agent_step = step


@PLAYER_REGISTRY.register('historical_player')
class HistoricalPlayer(Player):
    """Frozen snapshot; parent_id records which active player it came from."""

    _name = "HistoricalPlayer"

    def __init__(self, *args, parent_id: str = ''):
        super().__init__(*args)
        self._parent_id = parent_id

    @property
    def parent_id(self) -> str:
        return self._parent_id


class ActivePlayer(Player):
    """Trainable player: opponent selection + snapshot/mutate lifecycle."""

    _name = "ActivePlayer"
    BRANCH = namedtuple("BRANCH", ['name', 'prob'])

    def __init__(self, *args, **kwargs):
        super().__init__(*args)
        self._one_phase_step = int(float(self._cfg.get('one_phase_step', 2e5)))
        self._last_enough_step = 0
        self._strong_win_rate = self._cfg.get('strong_win_rate', 0.7)
        branch_probs = self._cfg.get('branch_probs', {'pfsp': 0.5, 'sp': 0.5})
        self._branch_probs = [self.BRANCH(k, v) for k, v in branch_probs.items()]

    def is_trained_enough(self, select_fn: Optional[Callable] = None) -> bool:
        if select_fn is None:
            select_fn = lambda p: isinstance(p, HistoricalPlayer)
        step_passed = self._total_agent_step - self._last_enough_step
        if step_passed < self._one_phase_step:
            return False
        elif step_passed >= 2 * self._one_phase_step:
            self._last_enough_step = self._total_agent_step
            return True
        else:
            historical = self._get_players(select_fn)
            if len(historical) == 0:
                return False
            win_rates = self._payoff[self, historical]
            if win_rates.min() > self._strong_win_rate:
                self._last_enough_step = self._total_agent_step
                return True
            return False

    def snapshot(self, metric_env=None) -> HistoricalPlayer:
        path = self.checkpoint_path.split('.pth')[0] + f'_{self._total_agent_step}' + '.pth'
        hp = HistoricalPlayer(
            self._cfg, self.category, self.payoff, path,
            self.player_id + f'_{int(self._total_agent_step)}', self._total_agent_step,
            parent_id=self.player_id
        )
        return hp

    def mutate(self, info: dict) -> Optional[str]:
        """Return a checkpoint path to reset to, or None to keep training."""
        return None

    def get_job(self, eval_flag: bool = False) -> dict:
        """Choose an opponent via the branch distribution."""
        p = np.random.uniform()
        total = sum(b.prob for b in self._branch_probs)
        acc = 0.0
        branch = self._branch_probs[-1].name
        for b in self._branch_probs:
            acc += b.prob / total
            if p < acc:
                branch = b.name
                break
        opponent = getattr(self, f'_{branch}_branch')()
        return {'opponent': opponent}

    def _get_players(self, select_fn: Callable) -> List[Player]:
        return [p for p in self._payoff.players if select_fn(p)]

    def _get_opponent(self, players: List[Player], p: Optional[np.ndarray] = None) -> Player:
        idx = np.random.choice(len(players), p=p)
        return players[idx]

    def _pfsp_branch(self) -> Player:
        historical = self._get_players(lambda p: isinstance(p, HistoricalPlayer))
        if not historical:
            return self
        win_rates = self._payoff[self, historical]
        p = pfsp(win_rates, weighting='squared')
        return self._get_opponent(historical, p)

    def _sp_branch(self) -> Player:
        return self

    def _verification_branch(self) -> Player:
        return self._pfsp_branch()


@PLAYER_REGISTRY.register('naive_sp_player')
class NaiveSpPlayer(ActivePlayer):
    """Simplest battle league player: PFSP vs snapshots when available,
    plain self-play otherwise (reference ding/league/player.py NaiveSpPlayer)."""
    _name = "NaiveSpPlayer"


@PLAYER_REGISTRY.register('main_player')
class MainPlayer(ActivePlayer):
    _name = "MainPlayer"


@PLAYER_REGISTRY.register('main_exploiter')
class MainExploiter(ActivePlayer):
    """Always plays the current main player; snapshots then resets."""

    _name = "MainExploiter"

    def _main_branch(self) -> Player:
        mains = self._get_players(lambda p: isinstance(p, MainPlayer))
        return self._get_opponent(mains) if mains else self

    def get_job(self, eval_flag: bool = False) -> dict:
        return {'opponent': self._main_branch()}

    def mutate(self, info: dict) -> Optional[str]:
        return info.get('pretrain_checkpoint_path')


@PLAYER_REGISTRY.register('league_exploiter')
class LeagueExploiter(ActivePlayer):
    """PFSP vs everyone; mutates back to the pretrained ckpt with prob 0.25."""

    _name = "LeagueExploiter"

    def mutate(self, info: dict) -> Optional[str]:
        if np.random.uniform() < 0.25:
            return info.get('pretrain_checkpoint_path')
        return None


def create_player(cfg: EasyDict, player_type: str, *args, **kwargs) -> Player:
    return PLAYER_REGISTRY.build(player_type, *args, **kwargs)
