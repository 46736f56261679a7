from .league import League
from .api import create_league_server
from .player import (Player, HistoricalPlayer, ActivePlayer, MainPlayer,
                     ExploiterPlayer, ExpertExploiterPlayer, MainExploiterPlayer,
                     ExpertPlayer, AdaptiveEvolutionaryExploiterPlayer)
from .algorithms import pfsp
from .payoff import Payoff
from .elo import ELORating, TrueSkill
