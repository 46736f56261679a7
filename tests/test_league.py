"""League control-plane tests: PFSP math, payoff warm-up, player branching,
snapshot/reset protocol, job dispatch and the HTTP API round trip."""
import numpy as np
import pytest
import torch

from distar_amd.league.algorithms import pfsp
from distar_amd.league.api import create_league_server
from distar_amd.league.elo import ELORating, TrueSkill
from distar_amd.league.league import League
from distar_amd.league.payoff import Payoff
from distar_amd.utils.config import Config
from distar_amd.utils.http import post_json


def test_pfsp_weightings():
    wr = np.array([0.1, 0.5, 0.9])
    for w in ('squared', 'variance', 'normal'):
        p = pfsp(wr, weighting=w)
        assert abs(p.sum() - 1) < 1e-8 and (p >= 0).all()
    sq = pfsp(wr, 'squared')
    assert sq[0] > sq[1] > sq[2]          # prioritize opponents we lose to
    var = pfsp(wr, 'variance')
    assert var[1] > var[0] and var[1] > var[2]
    assert np.allclose(pfsp(np.zeros(3)), np.full(3, 1 / 3))


def test_payoff_warmup_then_ema():
    p = Payoff(decay=0.9, warm_up_size=3, min_win_rate_games=2)
    assert p.win_rate_opponent('x') == 0.5    # no games yet
    for r in (1, 1, 0, 1):
        p.update('x', {'winrate': r, 'game_steps': 100, 'game_iters': 10,
                       'game_duration': 60})
    assert p.game_count['x'] == 4
    assert 0.5 < p.win_rate_opponent('x') < 1.0


def test_elo_and_trueskill():
    elo = ELORating()
    for _ in range(20):
        elo.update('a', 'b', 1)
    assert elo.ratings['a'] > elo.ratings['b']
    ts = TrueSkill()
    for _ in range(10):
        ts.update('a', 'b')
    assert ts.mu['a'] > ts.mu['b']
    assert ts.sigma['a'] < TrueSkill.SIGMA


def _league_cfg(tmp_path):
    return Config({
        'common': {'experiment_name': 'test_league'},
        'league': {
            'active_players': {
                'player_id': ['MP0', 'ME0', 'EP0'],
                'checkpoint_path': ['mp0.pth', 'me0.pth', 'ep0.pth'],
                'pipeline': ['default', 'default', 'default'],
                'frac_id': [1, 1, 1],
                'z_path': ['3map.json', '3map.json', '3map.json'],
                'z_prob': [0., 0., 0.],
                'teacher_id': ['sl', 'sl', 'sl'],
                'teacher_path': ['sl.pth', 'sl.pth', 'sl.pth'],
                'one_phase_step': [1000, 1000, 1000],
                'chosen_weight': [1.0, 1.0, 1.0],
            },
            'save_resume_freq': 100000,
        },
    })


@pytest.fixture
def league(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    lg = League(_league_cfg(tmp_path))
    yield lg
    lg.close()


def test_league_job_dispatch_and_results(league):
    from distar_amd.league.player import MainPlayer
    assert set(league.active_players) == {'MP0', 'ME0', 'EP0'}
    assert isinstance(league.active_players['MP0'], MainPlayer)
    # register + send train info: snapshot on phase completion
    resp = league.deal_with_register_learner({'player_id': 'MP0'})
    assert resp['ckpt_path'] == 'mp0.pth'
    resp = league.deal_with_learner_send_train_info(
        {'player_id': 'MP0', 'train_steps': 1500, 'checkpoint_path': 'mp0.pth'})
    assert resp['reset_checkpoint_path'] == 'none'   # MainPlayer never resets
    assert any(p.parent_id == 'MP0' for p in league.historical_players.values())
    # job dispatch now has historical players to sample
    for _ in range(8):
        job = league.deal_with_actor_ask_for_job({'job_type': 'train'})
        assert len(job['player_ids']) == 2
        assert job['env_info']['map_name'] == 'KingsCove'
        assert set(job) >= {'checkpoint_paths', 'pipelines', 'z_path',
                            'teacher_player_ids', 'send_data_players', 'branch'}
    # result ingestion updates payoff + elo
    league.deal_with_actor_send_result({
        'game_steps': 1000, 'game_iters': 100, 'game_duration': 600,
        '0': {'player_id': 'MP0', 'opponent_id': 'ME0', 'winloss': 1,
              'race_id': 1, 'z_type': 0},
        '1': {'player_id': 'ME0', 'opponent_id': 'MP0', 'winloss': -1,
              'race_id': 1, 'z_type': 0},
    })
    import time
    for _ in range(100):
        if league.elo.game_count:
            break
        time.sleep(0.05)
    assert league.elo.game_count == 1
    assert league.active_players['MP0'].payoff.game_count['ME0'] == 1


def test_main_exploiter_resets_to_teacher(league):
    resp = league.deal_with_learner_send_train_info(
        {'player_id': 'ME0', 'train_steps': 1500, 'checkpoint_path': 'me0.pth'})
    # MainExploiter always resets after snapshot; teacher ckpt doesn't exist
    # on disk in this test so the raw path comes back
    assert resp['reset_checkpoint_path'] == 'sl.pth'


def test_league_resume_roundtrip(league):
    league.deal_with_learner_send_train_info(
        {'player_id': 'MP0', 'train_steps': 1500, 'checkpoint_path': 'mp0.pth'})
    path = league.save_resume()
    cfg = _league_cfg(None)
    cfg.league.resume_path = path
    lg2 = League(cfg)
    try:
        assert set(lg2.active_players) == set(league.active_players)
        assert set(lg2.historical_players) == set(league.historical_players)
        assert lg2.active_players['MP0'].total_agent_step == 1500
    finally:
        lg2.close()


def test_league_http_api(league):
    server = create_league_server(league, host='127.0.0.1').start()
    try:
        url = f'http://127.0.0.1:{server.port}'
        resp = post_json(url + '/league/register_learner', {'player_id': 'MP0'})
        assert resp['ckpt_path'] == 'mp0.pth'
        league.deal_with_learner_send_train_info(
            {'player_id': 'MP0', 'train_steps': 1500, 'checkpoint_path': 'mp0.pth'})
        job = post_json(url + '/league/actor_ask_for_job', {'job_type': 'train'})
        assert 'player_ids' in job and len(job['player_ids']) == 2
        r = post_json(url + '/league/actor_send_result', {
            'game_steps': 10, 'game_iters': 1, 'game_duration': 5,
            '0': {'player_id': 'MP0', 'opponent_id': 'EP0', 'winloss': 1,
                  'race_id': 1, 'z_type': 0},
            '1': {'player_id': 'EP0', 'opponent_id': 'MP0', 'winloss': -1,
                  'race_id': 1, 'z_type': 0}})
        assert r['ok']
        elo = post_json(url + '/league/show_elo')
        assert 'ratings' in elo
    finally:
        server.stop()


def test_vs_bot_and_ladder_jobs(league):
    league.cfg.vs_bot = True
    league.cfg.bot_probs = [1, 1]
    job = league.deal_with_actor_ask_for_job({'job_type': 'train'})
    assert job['branch'] == 'train_bot'
    assert job['env_info']['player_ids'][1].startswith('bot')
    league.cfg.vs_bot = False
    # ladder needs historical players
    league.deal_with_learner_send_train_info(
        {'player_id': 'MP0', 'train_steps': 1500, 'checkpoint_path': 'mp0.pth'})
    league.deal_with_learner_send_train_info(
        {'player_id': 'MP0', 'train_steps': 1500, 'checkpoint_path': 'mp0.pth'})
    job = league.deal_with_actor_ask_for_job({'job_type': 'ladder'})
    assert job['branch'] == 'ladder'
    assert job['send_data_players'] == []
    assert len(job['player_ids']) == 2


def test_trueskill_updates_on_results(league):
    import time
    for _ in range(5):
        league.deal_with_actor_send_result({
            'game_steps': 10, 'game_iters': 1, 'game_duration': 5,
            '0': {'player_id': 'MP0', 'opponent_id': 'EP0', 'winloss': 1,
                  'race_id': 1, 'z_type': 0},
            '1': {'player_id': 'EP0', 'opponent_id': 'MP0', 'winloss': -1,
                  'race_id': 1, 'z_type': 0}})
    deadline = time.time() + 20
    while league.elo.game_count < 5 and time.time() < deadline:
        time.sleep(0.05)
    assert league.trueskill.mu['MP0'] > league.trueskill.mu['EP0']


def test_adaptive_evolutionary_reset_band(league):
    """AE exploiter resets to the hardest historical snapshot in the 20-50%
    win-rate band; teacher path when none qualify (reference player.py:640-760
    TStarBot-X policy)."""
    from distar_amd.league.player import AdaptiveEvolutionaryExploiterPlayer
    league.add_active_player(
        ckpt_path='ae.pth', pipeline='default', frac_id=1,
        z_path='3map.json', z_prob=0., teacher_id='sl',
        teacher_ckpt='sl_teacher.pth', one_phase_step=1000,
        chosen_weight=1.0, player_id='AE0')
    ae = league.active_players['AE0']
    assert isinstance(ae, AdaptiveEvolutionaryExploiterPlayer)
    assert ae.is_reset()            # always resets
    # empty historical pool -> teacher checkpoint
    assert ae.reset_checkpoint(league.active_players, {}, 'none') == 'sl_teacher.pth'
    # snapshot the mains, then shape win rates: H1 easy (0.9), H2 in-band (0.3)
    mp = league.active_players['MP0']
    h1, h2 = mp.snapshot(), mp.snapshot()
    hist = {h1.player_id: h1, h2.player_id: h2}
    def stat(w):
        return {'winrate': w, 'game_steps': 100, 'game_iters': 10,
                'game_duration': 60}
    ae.payoff._min_win_rate_games = 0          # skip the 0.5 warm-up gate
    ae.payoff.update(h1.player_id, stat(1.0))  # easy snapshot
    for w in [0, 0, 0, 0, 0, 0, 0, 1, 1, 1]:   # ~0.3 in-band
        ae.payoff.update(h2.player_id, stat(float(w)))
    wr = {pid: ae.payoff.pfsp_winrate_info_dict.get(pid, 0.5) for pid in hist}
    target = ae.reset_checkpoint(league.active_players, hist, 'none')
    in_band = [pid for pid, w in wr.items() if 0.2 <= w <= 0.5]
    if in_band:
        assert target in {hist[p].checkpoint_path for p in in_band}
    else:
        assert target == 'sl_teacher.pth'


def test_expert_exploiter_z_style_rotation(league):
    """EEP rotates among hand-picked style Zs on reset (reference
    player.py:425-525)."""
    from distar_amd.league.player import ExpertExploiterPlayer
    league.add_active_player(
        ckpt_path='eep.pth', pipeline='default', frac_id=1,
        z_path=['mutalisk.json', 'worker_rush.json', '3map.json'],
        z_prob=0., teacher_id='sl', teacher_ckpt='sl.pth',
        one_phase_step=1000, chosen_weight=1.0, player_id='EEP0')
    eep = league.active_players['EEP0']
    assert isinstance(eep, ExpertExploiterPlayer)
    assert eep.z_path in eep.z_paths
    seen = set()
    for _ in range(40):
        assert eep.is_reset()       # resets after every snapshot, rotating Z
        seen.add(eep.z_path)
    assert len(seen) >= 2           # rotation actually samples styles
    # snapshot names carry the style tag; reset target = newest MP snapshot
    hp = eep.snapshot()
    assert hp.player_id.startswith('EEP0H')
    mp = league.active_players['MP0']
    h1, h2 = mp.snapshot(), mp.snapshot()
    hist = {h1.player_id: h1, h2.player_id: h2}
    assert eep.reset_checkpoint(league.active_players, hist, 'none') == \
        h2.checkpoint_path


def test_phase_gate_snapshot_protocol(league):
    """total_agent_step crossing one_phase_step makes is_trained_enough
    true -> league snapshots into the historical pool (reference
    league.py:259-297)."""
    mp = league.active_players['MP0']
    assert not mp.is_trained_enough(league.historical_players,
                                    league.active_players)
    mp.total_agent_step = mp.one_phase_step + 1
    n_hist = len(league.historical_players)
    league.deal_with_learner_send_train_info(
        {'player_id': 'MP0', 'train_steps': 0, 'checkpoint_path': 'mp0.pth'})
    assert len(league.historical_players) >= n_hist


def test_league_debug_api_extended(league):
    """Debug/ops endpoints (reference league_api.py:56-305 subset)."""
    import requests
    from distar_amd.league.api import create_league_server
    api = create_league_server(league, host='127.0.0.1').start()
    url = f'http://127.0.0.1:{api.port}'
    try:
        r = requests.post(f'{url}/league/show_dist_stat', json={})
        assert 'MP0' in r.json()
        r = requests.post(f'{url}/league/show_trueskill', json={})
        assert 'mu' in r.json()
        r = requests.post(f'{url}/league/show_config', json={})
        assert 'branch_probs' in r.json()
        r = requests.post(f'{url}/league/display_player',
                          json={'player_id': 'MP0'})
        assert r.json()['MP0']['pipeline'] == 'default'
        r = requests.post(f'{url}/league/update_config',
                          json={'overrides': {'print_freq': 7}})
        assert r.json()['ok'] and league.cfg.print_freq == 7
        # snapshot then remove a historical player over the API
        mp = league.active_players['MP0']
        hp = mp.snapshot()
        league.set_hist_player(hp)
        r = requests.post(f'{url}/league/remove_hist_player',
                          json={'player_id': hp.player_id})
        assert r.json()['ok'] and hp.player_id not in league.historical_players
    finally:
        api.stop()


def test_league_resume_restores_payoff_and_ratings(league):
    """Resume restores deep state: payoff meters, ELO/TrueSkill ratings and
    snapshot lineage (reference league.py:535-556)."""
    mp = league.active_players['MP0']
    hp = mp.snapshot()
    league.set_hist_player(hp)
    def stat(w):
        return {'winrate': w, 'game_steps': 10, 'game_iters': 1,
                'game_duration': 5}
    for w in (1.0, 0.0, 1.0):
        mp.payoff.update('ME0', stat(w))
    league.elo.update('MP0', 'ME0', 1.0)
    league.trueskill.update('MP0', 'ME0')
    path = league.save_resume()
    cfg = _league_cfg(None)
    cfg.league.resume_path = path
    lg2 = League(cfg)
    try:
        mp2 = lg2.active_players['MP0']
        assert hp.player_id in lg2.historical_players
        assert lg2.historical_players[hp.player_id].parent_id == 'MP0'
        r1 = mp.payoff.stat_info_record['ME0']['winrate']
        r2 = mp2.payoff.stat_info_record['ME0']['winrate']
        assert r2.count == r1.count and abs(r2.val - r1.val) < 1e-9
        assert lg2.elo.ratings['MP0'] == league.elo.ratings['MP0']
        assert abs(lg2.trueskill.mu['MP0'] - league.trueskill.mu['MP0']) < 1e-9
    finally:
        lg2.close()


def test_warmup_ema_meter_phases():
    """Uniform mean during warm-up, EMA after (reference MoveAverageMeter
    semantics, log_helper.py:483-522)."""
    from distar_amd.league.meters import WarmupEmaMeter
    m = WarmupEmaMeter(decay=0.9, warm_up_size=3)
    for v in (1.0, 2.0, 3.0):
        m.update(v)
    assert abs(m.val - 2.0) < 1e-9          # plain mean of the warm-up
    m.update(12.0)                          # 0.9*2 + 0.1*12
    assert abs(m.val - 3.0) < 1e-9


def test_league_telemetry_stats():
    """DistStat/CumStat/UnitNumStat aggregate and pickle (they ride in the
    league resume file — reference {dist,cum,unit_num}_stat.py)."""
    import pickle
    from distar_amd.league.stats import CumStat, DistStat, UnitNumStat
    d = DistStat(decay=0.5, warm_up_size=1)
    d.update('zerg', {'dist/bo': 4.0, 'winloss': 1, 'player_id': 'MP0',
                      'text': 'ignored'})
    d.update('zerg', {'dist/bo': 8.0})
    # warm_up 1 -> first sets 4, second EMA: 0.5*4 + 0.5*8 = 6
    assert abs(d.stat_info_dict['zerg']['dist/bo'] - 6.0) < 1e-9
    assert d.game_count['zerg'] == 2

    c = CumStat(decay=0.5, warm_up_size=1)
    c.update('zerg', {'z_type': 2, 'spine': 1.0})
    c.update('zerg', {'z_type': 1, 'spine': 3.0})
    agg = c.stat_info_dict['zerg']['spine']
    assert agg[2] == 1.0 and agg[1] == 3.0 and agg[0] == 0.0

    u = UnitNumStat(decay=0.5, warm_up_size=1)
    u.update('zerg', 0, {'unit_num': {'drone': 12, 'zergling': 6}})
    assert u.stat_info_dict['zerg']['drone'] == 12.0
    # all three must survive the league resume pickle
    for obj in (d, c, u):
        clone = pickle.loads(pickle.dumps(obj))
        assert clone.stat_info_dict == obj.stat_info_dict
