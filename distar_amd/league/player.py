"""League player hierarchy (reference `ctools/worker/league/player.py`):

  Player -> HistoricalPlayer (fixed ckpt, parent_id)
         -> ActivePlayer     (one_phase_step, snapshot, reset, payoffs)
            -> MainPlayer                   (sp / pfsp / eval branches)
            -> ExploiterPlayer              (league exploiter, 25% reset)
            -> ExpertExploiterPlayer        (rotates hand-picked Z styles)
            -> MainExploiterPlayer          (vs_main branch, always resets)
            -> ExpertPlayer                 (fixed expert style)
            -> AdaptiveEvolutionaryExploiterPlayer
               (TStarBot-X style reset to the best 20-50%-win-rate snapshot)
"""
import pprint
import random
from abc import abstractmethod

import numpy as np

from .algorithms import pfsp
from .payoff import Payoff
from .stats import CumStat, DistStat, UnitNumStat

FRAC_ID = {0: ['zerg', 'terran', 'protoss'], 1: ['zerg'], 2: ['terran'], 3: ['protoss']}


class Player:
    _name = 'BasePlayer'
    _stat_keys = ['checkpoint_path', 'player_id', 'pipeline', 'frac_id', 'z_path',
                  'z_prob', 'teacher_id', 'teacher_checkpoint_path',
                  'total_agent_step', 'decay', 'warm_up_size',
                  'min_win_rate_games', 'total_game_count']
    _log_keys = ['payoff', 'teammate_payoff', 'opponent_payoff', 'dist_stat',
                 'cum_stat', 'unit_num_stat']

    def __init__(self, checkpoint_path, player_id, pipeline, frac_id, z_path,
                 z_prob, teacher_id, teacher_checkpoint_path,
                 total_agent_step=0, decay=0.99, warm_up_size=1000,
                 min_win_rate_games=200, total_game_count=0, payoff=None):
        self.checkpoint_path = checkpoint_path
        self.player_id = player_id
        self.pipeline = pipeline
        self.frac_id = frac_id
        self.z_path = z_path
        self.z_prob = z_prob
        self.teacher_id = teacher_id
        self.teacher_checkpoint_path = teacher_checkpoint_path
        self.total_agent_step = total_agent_step
        self.decay = decay
        self.warm_up_size = warm_up_size
        self.min_win_rate_games = min_win_rate_games
        self.total_game_count = total_game_count
        self.payoff = Payoff(decay, warm_up_size, min_win_rate_games)
        if payoff:
            self.payoff._stat_info_record = payoff._stat_info_record

    def get_race(self):
        return random.choice(FRAC_ID[self.frac_id])

    def reset_stats(self, stat_types=()):
        for k in (stat_types or self._log_keys):
            fn = getattr(self, f'reset_{k}', None)
            if fn is not None:
                fn()

    def reset_payoff(self):
        self.payoff = Payoff(self.decay, self.warm_up_size, self.min_win_rate_games)

    def __repr__(self):
        return pprint.pformat({k: getattr(self, k, None) for k in self._stat_keys})


class HistoricalPlayer(Player):
    _name = 'HistoricalPlayer'
    _stat_keys = Player._stat_keys + ['parent_id']

    def __init__(self, checkpoint_path, player_id, pipeline, frac_id, z_path,
                 z_prob, total_agent_step=0, decay=0.995, warm_up_size=1000,
                 min_win_rate_games=200, total_game_count=0, parent_id='none',
                 payoff=None):
        super().__init__(checkpoint_path, player_id, pipeline, frac_id, z_path,
                         z_prob, 'none', 'none', total_agent_step, decay,
                         warm_up_size, min_win_rate_games, total_game_count, payoff)
        self.parent_id = parent_id


def _hist_nonbot(historical_players, pfsp_train_bot):
    if pfsp_train_bot:
        return list(historical_players.keys())
    return [pid for pid, p in historical_players.items() if p.pipeline != 'bot']


class ActivePlayer(Player):
    _name = 'ActivePlayer'
    _stat_keys = Player._stat_keys + ['one_phase_step', 'chosen_weight',
                                      'last_enough_step', 'snapshot_times',
                                      'strong_win_rate']

    def __init__(self, checkpoint_path, player_id, pipeline, frac_id, z_path,
                 z_prob, teacher_id, teacher_checkpoint_path, chosen_weight=1.0,
                 total_agent_step=0, decay=0.995, warm_up_size=1000,
                 min_win_rate_games=200, total_game_count=0,
                 one_phase_step=int(2e8), last_enough_step=0, snapshot_times=0,
                 strong_win_rate=0.7, payoff=None, teammate_payoff=None,
                 opponent_payoff=None, dist_stat=None, cum_stat=None,
                 unit_num_stat=None, successive_model_path=None):
        super().__init__(checkpoint_path, player_id, pipeline, frac_id, z_path,
                         z_prob, teacher_id, teacher_checkpoint_path,
                         total_agent_step, decay, warm_up_size,
                         min_win_rate_games, total_game_count, payoff)
        self.one_phase_step = one_phase_step
        self.last_enough_step = last_enough_step
        self.snapshot_times = snapshot_times
        self.strong_win_rate = strong_win_rate
        self.snapshot_flag = False
        self.reset_flag = False
        self.chosen_weight = chosen_weight
        self.successive_model_path = successive_model_path or checkpoint_path
        self.last_successive_step = last_enough_step
        self.teammate_payoff = teammate_payoff or Payoff(decay, warm_up_size, min_win_rate_games)
        self.opponent_payoff = opponent_payoff or Payoff(decay, warm_up_size, min_win_rate_games)
        self.dist_stat = dist_stat or DistStat(decay, warm_up_size)
        self.cum_stat = cum_stat or CumStat(decay, warm_up_size)
        self.unit_num_stat = unit_num_stat or UnitNumStat(decay, warm_up_size)

    @abstractmethod
    def get_branch_opponent(self, historical_players, active_players,
                            branch_probs_dict, pfsp_train_bot=False):
        raise NotImplementedError

    def is_save_successive_model(self):
        if self.total_agent_step - self.last_successive_step > self.one_phase_step / 2:
            self.last_successive_step = self.total_agent_step
            return True
        return False

    def snapshot(self):
        self.snapshot_times += 1
        h_player_id = self.player_id + f'H{self.snapshot_times}'
        h_path = self.checkpoint_path.split('.pth')[0] + \
            f'_{self.total_agent_step}' + '.pth'
        return HistoricalPlayer(checkpoint_path=h_path, player_id=h_player_id,
                                pipeline=self.pipeline, frac_id=self.frac_id,
                                z_path=self.z_path, z_prob=self.z_prob,
                                total_agent_step=self.total_agent_step,
                                decay=self.decay, warm_up_size=self.warm_up_size,
                                min_win_rate_games=self.min_win_rate_games,
                                parent_id=self.player_id)

    def is_reset(self):
        return False

    # step-count + win-rate "trained enough" core shared by all actives
    def _phase_gate(self):
        if self.snapshot_flag:
            self.snapshot_flag = False
            self.last_enough_step = self.total_agent_step
            return True
        step_passed = self.total_agent_step - self.last_enough_step
        if step_passed < self.one_phase_step / 2:
            return False
        if step_passed >= self.one_phase_step:
            self.last_enough_step = self.total_agent_step
            return True
        return None     # undecided: check win rates

    def _beats_all(self, opponent_keys, margin=0.0):
        for pid in opponent_keys:
            rec = self.payoff.stat_info_record
            if pid not in rec:
                return False
            if not (rec[pid]['winrate'].val > self.strong_win_rate + margin and
                    rec[pid]['winrate'].count >= self.warm_up_size):
                return False
        return True

    def is_trained_enough(self, historical_players, active_players,
                          pfsp_train_bot=False, **kwargs):
        gate = self._phase_gate()
        if gate is not None:
            return gate
        hist_keys = _hist_nonbot(historical_players, pfsp_train_bot)
        if self._beats_all(hist_keys):
            self.last_enough_step = self.total_agent_step
            return True
        return False

    def reset_checkpoint(self, active_players, historical_players, new_player_id):
        return self.teacher_checkpoint_path

    def reset_teammate_payoff(self):
        self.teammate_payoff = Payoff(self.decay, self.warm_up_size, self.min_win_rate_games)

    def reset_opponent_payoff(self):
        self.opponent_payoff = Payoff(self.decay, self.warm_up_size, self.min_win_rate_games)

    def reset_dist_stat(self):
        self.dist_stat = DistStat(self.decay, self.warm_up_size)

    def reset_cum_stat(self):
        self.cum_stat = CumStat(self.decay, self.warm_up_size)

    def reset_unit_num_stat(self):
        self.unit_num_stat = UnitNumStat(self.decay, self.warm_up_size)

    def _pfsp_pick(self, historical_players, keys, weighting='squared'):
        weights = [self.payoff.pfsp_winrate_info_dict.get(pid, 0.5) for pid in keys]
        probs = pfsp(np.array(weights), weighting=weighting)
        pid = random.choices(keys, weights=probs, k=1)[0]
        return historical_players[pid]


class MainPlayer(ActivePlayer):
    _name = 'MainPlayer'

    def get_branch_opponent(self, historical_players, active_players,
                            branch_probs_dict, pfsp_train_bot=False):
        branch_probs = branch_probs_dict[self._name]
        branch = random.choices(list(branch_probs.keys()),
                                weights=list(branch_probs.values()), k=1)[0]
        if branch == 'sp':
            main_players = [p for p in active_players.values()
                            if isinstance(p, MainPlayer)]
            opponent = random.choice(main_players)
            # weak-opponent fallback: play that main's snapshots instead
            if opponent is not self and \
                    self.payoff.pfsp_winrate_info_dict.get(opponent.player_id, 0.5) < 0.3:
                keys = [pid for pid, p in historical_players.items()
                        if p.parent_id == opponent.player_id]
                if not keys:
                    keys = _hist_nonbot(historical_players, pfsp_train_bot=False)
                if keys:
                    opponent = self._pfsp_pick(historical_players, keys, 'variance')
            return branch, [self], [opponent]
        if branch == 'pfsp':
            keys = _hist_nonbot(historical_players, pfsp_train_bot)
            if not keys:        # before the first snapshot: self-play
                return 'sp', [self], [self]
            return branch, [self], [self._pfsp_pick(historical_players, keys, 'squared')]
        if branch == 'eval':
            if not historical_players:
                return 'sp', [self], [self]
            pid = random.choice(list(historical_players.keys()))
            return branch, [self], [historical_players[pid]]
        raise NotImplementedError(branch)

    def is_trained_enough(self, historical_players, active_players,
                          pfsp_train_bot=False, **kwargs):
        gate = self._phase_gate()
        if gate is not None:
            return gate
        hist_keys = _hist_nonbot(historical_players, pfsp_train_bot)
        # beats every snapshot by a margin, or beats everyone (incl. actives)
        if self._beats_all(hist_keys, margin=0.1):
            return True
        active_keys = [pid for pid in active_players if pid != self.player_id]
        if self._beats_all(hist_keys + active_keys):
            self.last_enough_step = self.total_agent_step
            return True
        return False


class ExploiterPlayer(ActivePlayer):
    _name = 'ExploiterPlayer'
    _reset_prob = 0.25

    def get_branch_opponent(self, historical_players, active_players,
                            branch_probs_dict, pfsp_train_bot=False):
        branch_probs = branch_probs_dict[self._name]
        branch = random.choices(list(branch_probs.keys()),
                                weights=list(branch_probs.values()), k=1)[0]
        mains = [p for p in active_players.values() if isinstance(p, MainPlayer)]
        if branch == 'pfsp':
            keys = _hist_nonbot(historical_players, pfsp_train_bot)
            if not keys and mains:  # before the first snapshot: vs active main
                return 'pfsp', [self], [random.choice(mains)]
            return branch, [self], [self._pfsp_pick(historical_players, keys, 'normal')]
        if branch == 'eval':
            if not historical_players and mains:
                return 'eval', [self], [random.choice(mains)]
            pid = random.choice(list(historical_players.keys()))
            return branch, [self], [historical_players[pid]]
        raise NotImplementedError(branch)

    def is_reset(self):
        if self.reset_flag:
            self.reset_flag = False
            return True
        return np.random.uniform() < self._reset_prob


class ExpertExploiterPlayer(ExploiterPlayer):
    """Rotates a hand-picked Z style on every reset; resets after every
    snapshot; reset target = newest main snapshot."""
    _name = 'ExpertExploiterPlayer'

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        assert isinstance(self.z_path, (list, tuple)), \
            'ExpertExploiterPlayer takes a list of style z paths'
        self.z_paths = list(self.z_path)
        self.z_path = random.choice(self.z_paths)

    def get_branch_opponent(self, historical_players, active_players,
                            branch_probs_dict, pfsp_train_bot=False):
        branch_probs = branch_probs_dict[self._name]
        branch = random.choices(list(branch_probs.keys()),
                                weights=list(branch_probs.values()), k=1)[0]
        mains = [p for p in active_players.values() if isinstance(p, MainPlayer)]
        if branch == 'pfsp':
            keys = _hist_nonbot(historical_players, pfsp_train_bot)
            if not keys and mains:  # before the first snapshot: vs active main
                return 'pfsp', [self], [random.choice(mains)]
            return branch, [self], [self._pfsp_pick(historical_players, keys, 'normal')]
        if branch == 'eval':
            if not historical_players and mains:
                return 'eval', [self], [random.choice(mains)]
            pid = random.choice(list(historical_players.keys()))
            return branch, [self], [historical_players[pid]]
        raise NotImplementedError(branch)

    def is_reset(self):
        self.z_path = random.choice(self.z_paths)
        return True

    def snapshot(self):
        hp = super().snapshot()
        style = str(self.z_path).split('.')[0]
        hp.player_id = self.player_id + f'H{self.snapshot_times}_{style}'
        return hp

    def reset_checkpoint(self, active_players, historical_players, new_player_id):
        mains = sorted((pid for pid in historical_players if 'MP' in pid),
                       key=lambda x: int(x.split('H')[-1].split('_')[0]))
        if mains:
            return historical_players[mains[-1]].checkpoint_path
        return self.teacher_checkpoint_path


class MainExploiterPlayer(ActivePlayer):
    _name = 'MainExploiterPlayer'

    def get_branch_opponent(self, historical_players, active_players,
                            branch_probs_dict, pfsp_train_bot=False):
        main_player_id = f'MP{self.player_id[-1]}'
        main_player = active_players[main_player_id]
        branch_probs = branch_probs_dict[self._name]
        branch = random.choices(list(branch_probs.keys()),
                                weights=list(branch_probs.values()), k=1)[0]
        if branch == 'vs_main':
            if self.payoff.pfsp_winrate_info_dict.get(main_player_id, 0.5) > 0.2:
                return branch, [self], [main_player]
            branch = 'pfsp'
        elif branch == 'eval':
            return 'vs_main_eval', [self], [main_player]
        if branch == 'pfsp':
            keys = [pid for pid, p in historical_players.items()
                    if p.parent_id == main_player_id]
            if not keys:
                return 'vs_main', [self], [main_player]
            return branch, [self], [self._pfsp_pick(historical_players, keys, 'variance')]
        raise NotImplementedError(branch)

    def is_trained_enough(self, historical_players, active_players,
                          pfsp_train_bot=False, **kwargs):
        gate = self._phase_gate()
        if gate is not None:
            return gate
        main_player_id = f'MP{self.player_id[-1]}'
        rec = self.payoff.stat_info_record
        if main_player_id in rec and \
                rec[main_player_id]['winrate'].val > self.strong_win_rate and \
                rec[main_player_id]['winrate'].count >= self.warm_up_size:
            self.last_enough_step = self.total_agent_step
            return True
        return False

    def is_reset(self):
        return True          # always restart from the teacher after snapshot


class ExpertPlayer(ActivePlayer):
    """Fixed expert style, trains pfsp vs history only, never resets."""
    _name = 'ExpertPlayer'

    def get_branch_opponent(self, historical_players, active_players,
                            branch_probs_dict, pfsp_train_bot=False):
        branch_probs = branch_probs_dict.get(self._name, {'pfsp': 1.0})
        branch = random.choices(list(branch_probs.keys()),
                                weights=list(branch_probs.values()), k=1)[0]
        keys = _hist_nonbot(historical_players, pfsp_train_bot)
        if branch == 'pfsp' and keys:
            return branch, [self], [self._pfsp_pick(historical_players, keys, 'normal')]
        pid = random.choice(list(historical_players.keys()))
        return 'eval', [self], [historical_players[pid]]


class AdaptiveEvolutionaryExploiterPlayer(ActivePlayer):
    """TStarBot-X style: on reset, restart from the historical snapshot whose
    win rate against this player is in the 20-50% band (hardest beatable)."""
    _name = 'AdaptiveEvolutionaryExploiterPlayer'

    def get_branch_opponent(self, historical_players, active_players,
                            branch_probs_dict, pfsp_train_bot=False):
        branch_probs = branch_probs_dict.get(self._name, {'pfsp': 1.0})
        branch = random.choices(list(branch_probs.keys()),
                                weights=list(branch_probs.values()), k=1)[0]
        keys = _hist_nonbot(historical_players, pfsp_train_bot)
        if branch == 'pfsp' and keys:
            return branch, [self], [self._pfsp_pick(historical_players, keys, 'variance')]
        pid = random.choice(list(historical_players.keys()))
        return 'eval', [self], [historical_players[pid]]

    def is_reset(self):
        return True

    def reset_checkpoint(self, active_players, historical_players, new_player_id):
        candidates = []
        for pid, p in historical_players.items():
            if p.pipeline == 'bot' or pid == new_player_id:
                continue
            wr = self.payoff.pfsp_winrate_info_dict.get(pid, 0.5)
            if 0.2 <= wr <= 0.5:
                candidates.append((wr, pid))
        if candidates:
            _, pid = min(candidates)        # hardest in-band snapshot
            return historical_players[pid].checkpoint_path
        return self.teacher_checkpoint_path
