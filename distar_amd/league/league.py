"""League manager: player registries, PFSP job dispatch, payoff/ELO result
ingestion, snapshotting, learner resets, periodic resume.

Functional parity with the reference's `ctools/worker/league/league.py:30-875`
(player registry init from config, `deal_with_register_learner`,
`deal_with_learner_send_train_info` (snapshot + reset protocol),
`deal_with_actor_ask_for_job` (train / vs_bot / ladder jobs, map sampling),
result-queue thread updating payoff/ELO/telemetry, hourly lz4-pickled
resume).  Serialization uses our zlib-backed codec; scalar telemetry goes to
the JSONL scalar sink instead of tensorboardX.
"""
import itertools
import os
import queue
import random

import numpy as np
import threading
import time
from collections import defaultdict
from shutil import copyfile

from .elo import ELORating, TrueSkill
from .player import (ActivePlayer, AdaptiveEvolutionaryExploiterPlayer,
                     ExpertExploiterPlayer, ExpertPlayer, ExploiterPlayer,
                     HistoricalPlayer, MainExploiterPlayer, MainPlayer)
from ..utils.config import Config, deep_merge_dicts
from ..utils.log import ScalarLogger, TextLogger
from ..utils.serialize import read_file, save_file

PLAYER_TYPES = {
    'MainPlayer': MainPlayer, 'MP': MainPlayer,
    'ExploiterPlayer': ExploiterPlayer, 'EP': ExploiterPlayer,
    'ExpertExploiterPlayer': ExpertExploiterPlayer, 'EEP': ExpertExploiterPlayer,
    'MainExploiterPlayer': MainExploiterPlayer, 'ME': MainExploiterPlayer,
    'ExpertPlayer': ExpertPlayer, 'EXP': ExpertPlayer,
    'AdaptiveEvolutionaryExploiterPlayer': AdaptiveEvolutionaryExploiterPlayer,
    'AE': AdaptiveEvolutionaryExploiterPlayer,
}

DEFAULT_LEAGUE_CFG = Config({
    'common': {'experiment_name': 'league_default'},
    'league': {
        'resume_path': '', 'save_resume_freq': 3600,
        'stat_decay': 0.999, 'stat_warm_up_size': 1000,
        'payoff_min_win_rate_games': 200, 'print_freq': 100,
        'pfsp_train_bot': False, 'vs_bot': False,
        'use_historical_players': False,
        'map_names': ['KingsCove'], 'map_id_weights': [1],
        'branch_probs': {
            'MainPlayer': {'sp': 0.35, 'pfsp': 0.5, 'eval': 0.15},
            'ExploiterPlayer': {'pfsp': 0.9, 'eval': 0.1},
            'ExpertExploiterPlayer': {'pfsp': 0.9, 'eval': 0.1},
            'MainExploiterPlayer': {'vs_main': 0.9, 'eval': 0.1},
            'ExpertPlayer': {'pfsp': 1.0},
            'AdaptiveEvolutionaryExploiterPlayer': {'pfsp': 1.0},
        },
        'active_players': {},
        'historical_players': {},
    },
    'learner': {'use_dapo': False},
})


class League:
    def __init__(self, cfg):
        self._whole_cfg = deep_merge_dicts(DEFAULT_LEAGUE_CFG, cfg)
        self.cfg = self._whole_cfg.league
        self._lock = threading.RLock()
        exp_dir = os.path.join('experiments', self._whole_cfg.common.experiment_name)
        self._path_model = os.path.abspath(os.path.join(exp_dir, 'league_models'))
        self.resume_dir = os.path.join(exp_dir, 'league_resume')
        os.makedirs(self._path_model, exist_ok=True)
        os.makedirs(self.resume_dir, exist_ok=True)
        self._logger = TextLogger(os.path.join(exp_dir, 'log'), name='league')
        self._scalars = ScalarLogger(os.path.join(exp_dir, 'log'), name='league')
        self._stat_decay = self.cfg.stat_decay
        self._stat_warm_up_size = self.cfg.stat_warm_up_size
        seed = self.cfg.get('seed')
        if seed is not None:
            # league job dispatch (branch/opponent/map draws) uses the global
            # RNG; seeding here makes league schedules reproducible
            random.seed(seed)
            np.random.seed(seed % (2 ** 32))
        self._payoff_min_win_rate_games = self.cfg.payoff_min_win_rate_games
        self.elo = ELORating()
        self.trueskill = TrueSkill()
        self.api_info = defaultdict(list)
        self._init_league()
        self.save_resume_freq = self.cfg.save_resume_freq
        self._result_queue = queue.Queue()
        self._stop = False
        threading.Thread(target=self._send_result_loop, daemon=True).start()
        threading.Thread(target=self._save_resume_thread, daemon=True).start()

    # ------------------------------------------------------------------ init
    def _init_league(self):
        if self.cfg.resume_path and os.path.isfile(self.cfg.resume_path):
            self.logger.info(f'loading league resume: {self.cfg.resume_path}')
            self._load_resume(self.cfg.resume_path)
            return
        self.active_players = {}
        self.historical_players = {}
        ap = self.cfg.active_players
        if ap:
            n = len(ap.player_id)
            for i in range(n):
                self.add_active_player(
                    ckpt_path=ap.checkpoint_path[i], pipeline=ap.pipeline[i],
                    frac_id=ap.frac_id[i], z_path=ap.z_path[i],
                    teacher_id=ap.teacher_id[i], teacher_ckpt=ap.teacher_path[i],
                    player_id=ap.player_id[i],
                    one_phase_step=int(float(ap.one_phase_step[i])),
                    chosen_weight=ap.chosen_weight[i], z_prob=ap.z_prob[i])
        if self.cfg.use_historical_players and self.cfg.historical_players:
            hp = self.cfg.historical_players
            ids = hp.get('player_id') or [f'SL{i}' for i in range(len(hp.checkpoint_path))]
            for i, pid in enumerate(ids):
                self.set_hist_player(HistoricalPlayer(
                    checkpoint_path=hp.checkpoint_path[i], player_id=pid,
                    pipeline=hp.get('pipeline', ['default'] * len(ids))[i],
                    frac_id=hp.get('frac_id', [1] * len(ids))[i],
                    z_path=hp.get('z_path', ['3map.json'] * len(ids))[i],
                    z_prob=hp.get('z_prob', [0.] * len(ids))[i],
                    decay=self._stat_decay, warm_up_size=self._stat_warm_up_size,
                    min_win_rate_games=self._payoff_min_win_rate_games,
                    parent_id='none'))

    def add_active_player(self, ckpt_path, pipeline, frac_id, z_path, teacher_id,
                          teacher_ckpt, player_id, one_phase_step,
                          chosen_weight=1.0, z_prob=0.):
        cls = None
        for prefix, c in PLAYER_TYPES.items():
            if player_id.startswith(prefix):
                cls = c
                break
        if cls is None:
            cls = MainPlayer
        player = cls(checkpoint_path=ckpt_path, player_id=player_id,
                     pipeline=pipeline, frac_id=frac_id, z_path=z_path,
                     z_prob=z_prob, teacher_id=teacher_id,
                     teacher_checkpoint_path=teacher_ckpt,
                     chosen_weight=chosen_weight, one_phase_step=one_phase_step,
                     decay=self._stat_decay, warm_up_size=self._stat_warm_up_size,
                     min_win_rate_games=self._payoff_min_win_rate_games)
        with self._lock:
            self.active_players[player_id] = player
        if isinstance(player, MainPlayer) and self.cfg.get('save_initial_snapshot', False):
            self.save_snapshot(player)
        return player

    def set_hist_player(self, hp):
        self.logger.info(f'add historical player: {hp.player_id}')
        with self._lock:
            self.historical_players[hp.player_id] = hp

    @property
    def all_players(self):
        merged = dict(self.historical_players)
        merged.update(self.active_players)
        return merged

    @property
    def logger(self):
        return self._logger

    # -------------------------------------------------------------- snapshot
    def save_snapshot(self, player):
        hp = player.snapshot()
        hp.checkpoint_path = os.path.join(
            self._path_model, hp.player_id + '_' +
            os.path.basename(str(player.checkpoint_path)))
        if os.path.isfile(str(player.checkpoint_path)):
            copyfile(player.checkpoint_path, hp.checkpoint_path)
        else:       # fake-model leagues carry symbolic paths
            hp.checkpoint_path = player.checkpoint_path
        self.set_hist_player(hp)
        self.logger.info(f'snapshot {player.player_id} -> {hp.player_id}')
        return hp.player_id

    def save_successive_model(self, player):
        if not self._whole_cfg.learner.get('use_dapo', False):
            return
        if not os.path.isfile(str(player.checkpoint_path)):
            return
        tmp = os.path.join('experiments', self._whole_cfg.common.experiment_name,
                           'successive_model', player.player_id)
        os.makedirs(tmp, exist_ok=True)
        path = os.path.join(tmp, os.path.basename(player.checkpoint_path))
        copyfile(player.checkpoint_path, path)
        player.successive_model_path = path
        player.last_successive_step = player.total_agent_step

    # --------------------------------------------------------------- learner
    def deal_with_register_learner(self, request_info):
        player_id = request_info['player_id']
        assert player_id in self.active_players, \
            f'{player_id} not in {list(self.active_players)}'
        self.api_info[player_id].append(
            (request_info.get('ip'), request_info.get('port'),
             request_info.get('rank', 0), request_info.get('world_size', 1)))
        self.logger.info(f'register learner: {player_id}')
        return {'ckpt_path': self.active_players[player_id].checkpoint_path}

    def deal_with_learner_send_train_info(self, request_info):
        player_id = request_info['player_id']
        player = self.active_players[player_id]
        with self._lock:
            player.total_agent_step += request_info['train_steps']
            player.checkpoint_path = request_info['checkpoint_path']
        reset_flag = player.reset_flag
        new_hp_id = None
        if player.is_save_successive_model():
            self.save_successive_model(player)
        if player.is_trained_enough(self.historical_players, self.active_players,
                                    pfsp_train_bot=self.cfg.pfsp_train_bot):
            new_hp_id = self.save_snapshot(player)
            reset_flag |= player.is_reset()
        if reset_flag:
            player.reset_flag = False
            with self._lock:
                player.reset_stats()
                new_ckpt = player.reset_checkpoint(self.active_players,
                                                   self.historical_players, new_hp_id)
                dst = os.path.join(self._path_model,
                                   f'{player.player_id}_ckpt.pth.tar')
                if os.path.isfile(str(new_ckpt)):
                    copyfile(new_ckpt, dst)
                    player.checkpoint_path = dst
                else:
                    player.checkpoint_path = new_ckpt
            self.save_successive_model(player)
            self.logger.info(f'reset {player_id} -> {player.checkpoint_path}')
            return {'reset_checkpoint_path': player.checkpoint_path}
        return {'reset_checkpoint_path': 'none'}

    # ----------------------------------------------------------------- actor
    def choose_active_player(self):
        ids = list(self.active_players.keys())
        weights = [self.active_players[i].chosen_weight for i in ids]
        return self.active_players[random.choices(ids, weights=weights, k=1)[0]]

    def deal_with_actor_ask_for_job(self, request_info):
        job_type = request_info.get('job_type', 'train')
        if job_type == 'ladder':
            branch, job_info = self._get_ladder_job_info()
        else:
            pinned = request_info.get('player_id')
            if pinned and pinned in self.active_players:
                # targeted job (test/ops facility beyond the reference's
                # weighted sampling): train THIS active player
                player = self.active_players[pinned]
            else:
                player = self.choose_active_player()
            if self.cfg.get('vs_bot', False):
                branch, job_info = self._get_vs_bot_job_info(player)
            else:
                branch, job_info = self._get_train_job_info(player)
        map_name = random.choices(self.cfg.map_names,
                                  weights=self.cfg.map_id_weights, k=1)[0]
        job_info['env_info']['map_name'] = map_name
        job_info['branch'] = branch
        return job_info

    def _job_from_players(self, players, branch):
        successive = [p.player_id if isinstance(p, MainPlayer) else 'none'
                      for p in players]
        successive_paths = [p.successive_model_path if isinstance(p, MainPlayer)
                            else 'none' for p in players]
        job_info = {
            'player_ids': [p.player_id for p in players],
            'side_ids': list(range(len(players))),
            'pipelines': [p.pipeline for p in players],
            'checkpoint_paths': [p.checkpoint_path for p in players],
            'successive_ids': successive,
            'successive_model_paths': successive_paths,
            'z_path': [p.z_path for p in players],
            'z_prob': [p.z_prob for p in players],
            'teacher_player_ids': [p.teacher_id for p in players],
            'teacher_checkpoint_paths': [p.teacher_checkpoint_path for p in players],
            'send_data_players': list({p.player_id for p in players
                                       if isinstance(p, ActivePlayer)}),
            'update_players': list({p.player_id for p in players
                                    if isinstance(p, ActivePlayer)}),
            'frac_ids': [p.frac_id for p in players],
            'env_info': {'player_ids': [p.player_id for p in players],
                         'side_id': [0, 1]},
        }
        if branch == 'vs_main':
            for idx, p in enumerate(players):
                if isinstance(p, MainPlayer):
                    job_info['teacher_player_ids'][idx] = 'none'
                    job_info['teacher_checkpoint_paths'][idx] = 'none'
            job_info['send_data_players'] = \
                [p.player_id for p in players
                 if isinstance(p, ActivePlayer) and not isinstance(p, MainPlayer)]
        elif 'eval' in branch:
            job_info['teacher_player_ids'] = ['none'] * len(players)
            job_info['teacher_checkpoint_paths'] = ['none'] * len(players)
            job_info['send_data_players'] = []
        return job_info

    def _get_train_job_info(self, player):
        branch, home_team, opponent_team = player.get_branch_opponent(
            self.historical_players, self.active_players, self.cfg.branch_probs,
            self.cfg.get('pfsp_train_bot', False))
        players = list(itertools.chain.from_iterable(zip(opponent_team, home_team)))
        return branch, self._job_from_players(players, branch)

    def _get_vs_bot_job_info(self, player):
        bot_probs = self.cfg.get('bot_probs', [1] * 10)
        bot_level = random.choices(range(len(bot_probs)), weights=bot_probs, k=1)[0]
        bot_race = self.cfg.get('frac_id', 1)
        job_info = {
            'player_ids': [player.player_id], 'side_ids': [0],
            'checkpoint_paths': [player.checkpoint_path],
            'successive_ids': [player.player_id if isinstance(player, MainPlayer)
                               else 'none'],
            'pipelines': [player.pipeline], 'z_path': [player.z_path],
            'z_prob': [player.z_prob],
            'teacher_player_ids': [player.teacher_id],
            'teacher_checkpoint_paths': [player.teacher_checkpoint_path],
            'send_data_players': [player.player_id],
            'update_players': [player.player_id],
            'frac_ids': [player.frac_id, bot_race],
            'bot_id': f'bot{bot_level}',
            'env_info': {'player_ids': [player.player_id, f'bot{bot_level}'],
                         'side_id': [0, 1]},
        }
        return 'train_bot', job_info

    def _get_ladder_job_info(self):
        """Eval-only ELO round robin over historical players (+ ladder bots)."""
        less, enough = [], []
        hist = list(self.historical_players.values())
        ladder_bots = self.cfg.get('ladder_bots', [])
        candidates = hist + list(ladder_bots)
        min_games = self.cfg.get('ladder_min_games', 100)
        for home in candidates:
            for away in candidates:
                home_id = home if isinstance(home, str) else home.player_id
                away_id = away if isinstance(away, str) else away.player_id
                if 'bot' in home_id or home_id == away_id:
                    continue
                pair = [home, away, home_id, away_id]
                (less if self.elo.games[home_id][away_id] < min_games
                 else enough).append(pair)
        players = random.choice(less or enough)
        if isinstance(players[1], str):     # vs bot
            pipelines = [players[0].pipeline, players[1]]
            n = 1
        else:
            pipelines = [p.pipeline for p in players[:2]]
            n = 2
        job_info = {
            'player_ids': players[2:], 'side_ids': [0, 1],
            'pipelines': pipelines,
            'checkpoint_paths': [p.checkpoint_path for p in players[:n]],
            'successive_ids': ['none'] * n,
            'z_path': [p.z_path for p in players[:n]],
            'z_prob': [p.z_prob for p in players[:n]],
            'teacher_player_ids': ['none'] * n,
            'teacher_checkpoint_paths': ['none'] * n,
            'send_data_players': [], 'update_players': [],
            'frac_ids': [1, 1],
            'env_info': {'player_ids': players[2:], 'side_id': [0, 1]},
        }
        return 'ladder', job_info

    def deal_with_actor_send_result(self, request_info):
        self._result_queue.put(request_info)
        return True

    def _send_result_loop(self):
        while not self._stop:
            try:
                request_info = self._result_queue.get(timeout=0.1)
            except queue.Empty:
                continue
            try:
                self._ingest_result(request_info)
            except Exception as e:  # noqa: BLE001 - keep the loop alive
                self.logger.info(f'result ingestion error: {e!r}')

    def _ingest_result(self, request_info):
        game_steps = request_info.pop('game_steps', 0)
        game_iters = request_info.pop('game_iters', 0)
        game_duration = request_info.pop('game_duration', 0)
        sides = {k: v for k, v in request_info.items() if k in ('0', '1', 0, 1)}
        for side in sides.values():
            player_id = side['player_id']
            if player_id not in self.all_players:
                continue
            player = self.all_players[player_id]
            info = {'winrate': (1 + side['winloss']) / 2,
                    'game_steps': game_steps, 'game_iters': game_iters,
                    'game_duration': game_duration}
            with self._lock:
                if player_id != side['opponent_id']:
                    player.payoff.update(opponent_id=side['opponent_id'],
                                         stat_info=info)
                player.total_game_count += 1
        first = sides.get('0') or sides.get(0)
        if first is not None:
            with self._lock:
                self.elo.update(first['player_id'], first['opponent_id'],
                                first['winloss'])
                if first['winloss'] > 0:
                    self.trueskill.update(first['player_id'], first['opponent_id'])
                elif first['winloss'] < 0:
                    self.trueskill.update(first['opponent_id'], first['player_id'])
            if self.elo.game_count % 100 == 0:
                self.logger.info(self.elo.elo_text())
        for side in sides.values():
            player_id = side['player_id']
            player = self.all_players.get(player_id)
            if isinstance(player, ActivePlayer):
                frac_id = side.get('race_id', 0)
                with self._lock:
                    player.dist_stat.update(frac_id, side)
                    player.cum_stat.update(frac_id, side)
                    player.unit_num_stat.update(frac_id, side.get('side_id', 0),
                                                side)
                if player.total_game_count % self.cfg.print_freq == 0:
                    for opp, info in player.payoff.stat_info_dict.items():
                        for k, v in info.items():
                            self._scalars.add_scalar(f'{player_id}/{k}/{opp}', v,
                                                     player.total_game_count)
                    self.logger.info('=' * 20 + player_id + '=' * 20 + '\n' +
                                     player.payoff.get_text())

    # ---------------------------------------------------------------- resume
    def save_resume(self):
        path = os.path.join(self.resume_dir,
                            f'league_resume_{int(time.time())}.pkl')
        state = {'active_players': self.active_players,
                 'historical_players': self.historical_players,
                 'elo': self.elo, 'trueskill': self.trueskill}
        with self._lock:
            save_file(path, state, fs_type='nppickle')
        self.logger.info(f'saved league resume: {path}')
        return path

    def _load_resume(self, path):
        state = read_file(path, fs_type='nppickle')
        self.active_players = state['active_players']
        self.historical_players = state['historical_players']
        self.elo = state['elo']
        self.trueskill = state.get('trueskill', TrueSkill())

    def _save_resume_thread(self):
        last = time.time()
        while not self._stop:
            time.sleep(1)
            if time.time() - last >= self.save_resume_freq:
                last = time.time()
                try:
                    self.save_resume()
                except Exception as e:  # noqa: BLE001
                    self.logger.info(f'resume save failed: {e!r}')

    def close(self):
        self._stop = True
