"""Per-opponent match statistics (reference `ctools/worker/league/payoff.py`):
decayed winrate/game_steps/game_iters/game_duration per opponent, returning
0.5 until ``min_win_rate_games`` games are recorded."""
from collections import defaultdict

from .meters import WarmupEmaMeter


class Payoff:
    data_keys = ['winrate', 'game_steps', 'game_iters', 'game_duration']

    def __init__(self, decay=0.999, warm_up_size=1000, min_win_rate_games=1000):
        self._decay = decay
        self._warm_up_size = warm_up_size
        self._min_win_rate_games = min_win_rate_games
        self._stat_info_record = defaultdict(self._template)

    def _template(self):
        return {k: WarmupEmaMeter(self._decay, self._warm_up_size)
                for k in self.data_keys}

    def win_rate_opponent(self, opponent_id, min_win_rate_games=True):
        rec = self._stat_info_record[opponent_id]
        if min_win_rate_games and rec['winrate'].count < self._min_win_rate_games:
            return 0.5
        return rec['winrate'].val

    def update(self, opponent_id, stat_info):
        for k in self.data_keys:
            self._stat_info_record[opponent_id][k].update(stat_info[k])
        return True

    @property
    def pfsp_winrate_info_dict(self):
        return {p: self.win_rate_opponent(p) for p in self._stat_info_record}

    @property
    def stat_info_dict(self):
        return {opp: {k: rec[k].val for k in self.data_keys}
                for opp, rec in self._stat_info_record.items()}

    @property
    def stat_info_record(self):
        return self._stat_info_record

    @property
    def game_count(self):
        return {opp: rec['winrate'].count for opp, rec in self._stat_info_record.items()}

    def get_text(self):
        lines = [f"{'opponent':<28s} " + ' '.join(f'{k:>14s}' for k in self.data_keys)
                 + f" {'games':>8s}"]
        for opp, rec in sorted(self._stat_info_record.items()):
            lines.append(f'{opp:<28s} ' + ' '.join(f'{rec[k].val:14.4f}' for k in self.data_keys)
                         + f" {rec['winrate'].count:8d}")
        return '\n'.join(lines)
