"""Rating systems for eval ladders (reference `ctools/worker/ladder/elo.py`
and `trueskill_algo.py`): K-factor ELO with per-pair game counts, and a
self-contained 1-vs-1 TrueSkill update (Gaussian skill/beta model with the
standard v/w truncation corrections)."""
import math
from collections import Counter, defaultdict


class _InitRating(dict):
    """Picklable default-dict for ratings (league resume pickles this)."""

    def __init__(self, init_rating):
        super().__init__()
        self.init_rating = init_rating

    def __missing__(self, key):
        self[key] = float(self.init_rating)
        return self[key]


class _PairGames(dict):
    def __missing__(self, key):
        self[key] = Counter()
        return self[key]


class ELORating:
    def __init__(self, init_rating=1000., k_factor=16.):
        self.ratings = _InitRating(init_rating)
        self.games = _PairGames()
        self.game_count = 0

    @staticmethod
    def expect(ra, rb):
        return 1.0 / (1 + 10 ** ((rb - ra) / 400.))

    def update(self, home_id, away_id, winloss):
        """winloss: +1 home win, 0 draw, -1 home loss."""
        score = (1 + float(winloss)) / 2
        ra, rb = self.ratings[home_id], self.ratings[away_id]
        ea = self.expect(ra, rb)
        self.ratings[home_id] = ra + 16. * (score - ea)
        self.ratings[away_id] = rb + 16. * ((1 - score) - (1 - ea))
        self.games[home_id][away_id] += 1
        self.games[away_id][home_id] += 1
        self.game_count += 1

    def elo_text(self):
        lines = [f'{"player":<32s} {"elo":>8s} {"games":>7s}']
        for pid, r in sorted(self.ratings.items(), key=lambda kv: -kv[1]):
            n = sum(self.games[pid].values())
            lines.append(f'{pid:<32s} {r:8.1f} {n:7d}')
        return '\n'.join(lines)


class TrueSkill:
    """Minimal 2-player TrueSkill (mu=25, sigma=25/3, beta=sigma/2)."""

    MU, SIGMA = 25.0, 25.0 / 3

    def __init__(self):
        self.mu = _InitRating(self.MU)
        self.sigma = _InitRating(self.SIGMA)

    @staticmethod
    def _pdf(x):
        return math.exp(-x * x / 2) / math.sqrt(2 * math.pi)

    @staticmethod
    def _cdf(x):
        return 0.5 * (1 + math.erf(x / math.sqrt(2)))

    def update(self, winner, loser):
        beta = self.SIGMA / 2
        mu_w, mu_l = self.mu[winner], self.mu[loser]
        s_w, s_l = self.sigma[winner], self.sigma[loser]
        c2 = 2 * beta ** 2 + s_w ** 2 + s_l ** 2
        c = math.sqrt(c2)
        t = (mu_w - mu_l) / c
        denom = max(self._cdf(t), 1e-12)
        v = self._pdf(t) / denom
        w = v * (v + t)
        self.mu[winner] = mu_w + (s_w ** 2 / c) * v
        self.mu[loser] = mu_l - (s_l ** 2 / c) * v
        self.sigma[winner] = math.sqrt(max(s_w ** 2 * (1 - (s_w ** 2 / c2) * w), 1e-6))
        self.sigma[loser] = math.sqrt(max(s_l ** 2 * (1 - (s_l ** 2 / c2) * w), 1e-6))

    def rating(self, player):
        """Conservative skill estimate (mu - 3 sigma)."""
        return self.mu[player] - 3 * self.sigma[player]
