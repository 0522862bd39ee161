from .player import Player, ActivePlayer, HistoricalPlayer, MainPlayer, MainExploiter, LeagueExploiter, create_player
from .algorithm import pfsp, uniform
from .metric import EloCalculator, TrueSkillCalculator, PlayerRating, LeagueMetricEnv, get_elo, get_elo_array
from .shared_payoff import BattleSharedPayoff, create_payoff
from .base_league import BaseLeague, OneVsOneLeague, create_league
