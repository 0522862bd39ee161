"""Payoff table: win/draw/loss records between players.

Parity: reference ding/league/shared_payoff.py (BattleSharedPayoff) +
battle record counting.
"""
import copy
from collections import defaultdict
from typing import List, Optional, Union

import numpy as np

from ding.utils import EasyDict, LockContext, LockContextType


class BattleRecordDict(dict):

    data_keys = ['wins', 'draws', 'losses', 'games']

    def __init__(self):
        super().__init__()
        for k in self.data_keys:
            self[k] = 0


class BattleSharedPayoff:
    """Win-rate table keyed by (home_id, away_id)."""

    def __init__(self, cfg: EasyDict = None):
        self._cfg = cfg or EasyDict({})
        self._decay = self._cfg.get('decay', 0.99)
        self._min_win_rate_games = self._cfg.get('min_win_rate_games', 8)
        self._players = {}
        self._data = defaultdict(BattleRecordDict)
        self._lock = LockContext(LockContextType.THREAD_LOCK)

    @property
    def players(self) -> List:
        return list(self._players.values())

    def add_player(self, player) -> None:
        with self._lock:
            self._players[player.player_id] = player

    def update(self, job_info: dict) -> bool:
        """job_info: {'player_id': [home, away], 'result': [[...episode
        results 'wins'/'draws'/'losses']]}"""
        with self._lock:
            home, away = job_info['player_id']
            key = (home, away)
            for episode in job_info['result']:
                for result in (episode if isinstance(episode, list) else [episode]):
                    for k in BattleRecordDict.data_keys[:3]:
                        self._data[key][k] *= self._decay
                    self._data[key]['games'] *= self._decay
                    assert result in ('wins', 'draws', 'losses'), result
                    self._data[key][result] += 1
                    self._data[key]['games'] += 1
            return True

    def __getitem__(self, players: tuple) -> np.ndarray:
        """payoff[home, away(s)] -> win rate(s) of home vs away."""
        home, away = players
        if not isinstance(away, list):
            away = [away]
            single = True
        else:
            single = False
        rates = np.array([self._win_rate(home.player_id, a.player_id) for a in away])
        return rates[0] if single else rates

    def _win_rate(self, home_id: str, away_id: str) -> float:
        key, rkey = (home_id, away_id), (away_id, home_id)
        handle = self._data[key]
        reverse = self._data[rkey]
        games = handle['games'] + reverse['games']
        if games < self._min_win_rate_games:
            return 0.5
        wins = handle['wins'] + reverse['losses'] + 0.5 * (handle['draws'] + reverse['draws'])
        return wins / max(games, 1e-8)


def create_payoff(cfg: EasyDict) -> BattleSharedPayoff:
    return BattleSharedPayoff(cfg)
