"""End-to-end RL stack on CPU: coordinator + league(API) + actor (mock env)
+ RL learner with comm hooks, exchanging real trajectories/models through the
Adapter transport — the single-node wiring of BASELINE config 5."""
import threading
import time

import pytest
import torch

from distar_amd.actor.actor import Actor
from distar_amd.actor.comm import LearnerComm
from distar_amd.data.coordinator import Coordinator
from distar_amd.league.api import create_league_server
from distar_amd.league.league import League
from distar_amd.learner.rl_learner import RLLearner
from distar_amd.utils.config import Config


@pytest.mark.timeout(900)
def test_rl_league_loop(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    torch.manual_seed(0)
    coord = Coordinator().run()
    league = League(Config({
        'common': {'experiment_name': 'e2e'},
        'league': {
            'save_resume_freq': 10000,
            'active_players': {
                'player_id': ['MP0'], 'checkpoint_path': ['none'],
                'pipeline': ['default'], 'frac_id': [1],
                'z_path': ['3map.json'], 'z_prob': [0.0],
                'teacher_id': ['sl'], 'teacher_path': ['none'],
                'one_phase_step': [int(1e9)], 'chosen_weight': [1.0],
            }}}))
    api = create_league_server(league, host='127.0.0.1').start()
    cfg = Config({
        'common': {'experiment_name': 'e2e', 'type': 'train'},
        'communication': {'coordinator_ip': '127.0.0.1',
                          'coordinator_port': coord.port,
                          'league_ip': '127.0.0.1', 'league_port': api.port,
                          'adapter_traj_worker_num': 1,
                          'learner_send_model_freq': 1,
                          'learner_send_train_info_freq': 1},
        'env': {'player_num': 2, 'max_episode_steps': 100000},
        'actor': {'episode_num': -1, 'traj_len': 3, 'env_type': 'mock',
                  'job_type': 'train', 'use_cuda': False},
        'learner': {'player_id': 'MP0', 'job_type': 'train', 'use_cuda': False,
                    'use_amp': False,
                    'data': {'batch_size': 2, 'trajectory_length': 3,
                             'buffer_size': 2, 'use_async_cuda': False},
                    'hook': {'after_iter': {
                        'log_show': {'ext_args': {'freq': 1000}}}}},
        'model': {'enable_baselines': ['winloss', 'build_order',
                                       'built_unit', 'battle']},
    })
    actor = Actor(cfg)
    actor_thread = threading.Thread(target=actor.run, daemon=True)
    actor_thread.start()
    try:
        learner = RLLearner(cfg)
        comm = LearnerComm(cfg)
        resp = comm.register_learner(learner)
        assert 'ckpt_path' in resp
        learner._setup_comm_hooks(comm)
        learner.run(max_iterations=2)
        assert learner.last_iter.val == 2
        assert 'total_loss' in learner.record.var_dict
        # model was published for actors; league got train info
        assert comm.adapter.length('MP0model') >= 0
        deadline = time.time() + 30
        while league.active_players['MP0'].total_agent_step == 0 and \
                time.time() < deadline:
            time.sleep(0.5)
        assert league.active_players['MP0'].total_agent_step > 0
    finally:
        actor.close()
        actor_thread.join(timeout=60)
        learner._dataloader.close()
        league.close()
        api.stop()
        coord.close()


@pytest.mark.timeout(1800)
def test_three_player_league_loop(tmp_path, monkeypatch):
    """Config-5 league shape: MP0 (main) + ME0 (main exploiter) + EP0
    (exploiter) sharing one coordinator+league; each player's learner runs 1
    iteration off trajectories produced by a shared actor fleet (reference
    league defaults, `distar/bin/league.py` + user_config active_players)."""
    monkeypatch.chdir(tmp_path)
    torch.manual_seed(0)
    coord = Coordinator().run()
    n = 3
    ids = ['MP0', 'ME0', 'EP0']
    league = League(Config({
        'common': {'experiment_name': 'e2e3'},
        'league': {
            'save_resume_freq': 10000,
            # force data-producing branches.  NOTE: deep_merge keeps sibling
            # keys from the defaults, so 'eval' must be zeroed explicitly —
            # an actor that draws an eval job sends no training data and
            # holds the job for actor_ask_for_job_interval.
            'branch_probs': {
                'MainPlayer': {'sp': 1.0, 'pfsp': 0.0, 'eval': 0.0},
                'ExploiterPlayer': {'pfsp': 1.0, 'eval': 0.0},
                'MainExploiterPlayer': {'vs_main': 1.0, 'eval': 0.0}},
            'active_players': {
                'player_id': ids, 'checkpoint_path': ['none'] * n,
                'pipeline': ['default'] * n, 'frac_id': [1] * n,
                'z_path': ['3map.json'] * n, 'z_prob': [0.0] * n,
                'teacher_id': ['sl'] * n, 'teacher_path': ['none'] * n,
                'one_phase_step': [int(1e9)] * n, 'chosen_weight': [1.0] * n,
            }}}))
    api = create_league_server(league, host='127.0.0.1').start()

    def make_cfg(player_id):
        return Config({
            'common': {'experiment_name': 'e2e3', 'type': 'train'},
            'communication': {'coordinator_ip': '127.0.0.1',
                              'coordinator_port': coord.port,
                              'league_ip': '127.0.0.1', 'league_port': api.port,
                              'adapter_traj_worker_num': 1,
                              'learner_send_model_freq': 1,
                              'learner_send_train_info_freq': 1},
            'env': {'player_num': 2, 'max_episode_steps': 24},
            'actor': {'episode_num': -1, 'traj_len': 3, 'env_type': 'mock',
                      'job_type': 'train', 'use_cuda': False},
            'learner': {'player_id': player_id, 'job_type': 'train',
                        'use_cuda': False, 'use_amp': False,
                        'data': {'batch_size': 2, 'trajectory_length': 3,
                                 'buffer_size': 2, 'use_async_cuda': False},
                        'hook': {'after_iter': {
                            'log_show': {'ext_args': {'freq': 1000}}}}},
            'model': {'enable_baselines': ['winloss']},
        })

    # learners first so models are published before actors ask for jobs
    learners, comms = [], []
    for pid in ids:
        learner = RLLearner(make_cfg(pid))
        comm = LearnerComm(make_cfg(pid))
        comm.register_learner(learner)
        learner._setup_comm_hooks(comm)
        learners.append(learner)
        comms.append(comm)
    actors = []
    for pid in ids:
        acfg = make_cfg(pid)
        acfg.actor.job_player_id = pid      # pin each actor to one player
        actors.append(Actor(acfg))
    threads = [threading.Thread(target=a.run, daemon=True) for a in actors]
    for t in threads:
        t.start()
    try:
        done = {}

        def run_learner(pid, learner):
            learner.run(max_iterations=1)
            done[pid] = learner.last_iter.val

        lthreads = [threading.Thread(target=run_learner, args=(pid, ln),
                                     daemon=True)
                    for pid, ln in zip(ids, learners)]
        for t in lthreads:
            t.start()
        deadline = time.time() + 1200       # generous: loaded CI hosts
        while len(done) < n and time.time() < deadline:
            time.sleep(1)
        if len(done) < n:
            # diagnostics: where is the stuck learner, what do the queues say
            import faulthandler
            import sys
            for pid, ln in zip(ids, learners):
                if pid not in done:
                    q = ln._dataloader._batch_queue.qsize() \
                        if hasattr(ln._dataloader, '_batch_queue') else '?'
                    print(f'[diag] {pid}: batch_queue={q} '
                          f'adapter_len={ln._dataloader.adapter.length(pid + "traj")}',
                          flush=True)
            print(f'[diag] coordinator queues: '
                  f'{ {t: len(qq) for t, qq in coord._queues.items()} }', flush=True)
            faulthandler.dump_traceback(file=sys.stdout)
        assert done == {pid: 1 for pid in ids}, f'learners finished: {done}'
        # league saw game results from the actor fleet
        deadline = time.time() + 180
        def games():
            return sum(p.total_game_count for p in league.active_players.values())
        while games() == 0 and time.time() < deadline:
            time.sleep(0.5)
        assert games() > 0
    finally:
        for a in actors:
            a.close()
        for t in threads:
            t.join(timeout=60)
        for ln in learners:
            ln._dataloader.close()
        league.close()
        api.stop()
        coord.close()
