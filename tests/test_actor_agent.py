"""Agent + mock-env rollout integration: trajectories produced by
`Agent.collect_data` must collate through the RL pipeline into a valid
`rl_learner_forward` batch (the full actor->learner data contract)."""
import pytest
import torch

from distar_amd.actor.actor import Actor
from distar_amd.actor.agent import Agent
from distar_amd.envs.mock_env import MockSC2Env
from distar_amd.lib.fake_data import rl_collate
from distar_amd.losses import ReinforcementLoss
from distar_amd.models import Model
from distar_amd.utils.config import Config


@pytest.mark.timeout(600)
def test_agent_rollout_feeds_rl_learner():
    torch.manual_seed(0)
    cfg = Config({'common': {'type': 'train'},
                  'actor': {'traj_len': 3, 'job_type': 'train'},
                  'env': {'player_num': 2, 'max_episode_steps': 100000},
                  'agent': {}})
    env = MockSC2Env(cfg, entity_num_range=(24, 48), seed=1)
    agents = [Agent(cfg, env_id=0) for _ in range(2)]
    obs = env.reset()
    for i, agent in enumerate(agents):
        agent.player_id = f'MP{i}'
        agent.reset(obs=obs.get(i))
    trajs = []
    done = False
    last_obs = obs
    steps = 0
    while not done and steps < 12:
        actions = {i: agents[i].step(last_obs[i])[0] for i in agents and last_obs}
        for i, a in actions.items():
            assert set(a) >= {'func_id', 'skip_steps', 'queued', 'unit_tags',
                              'target_unit_tag', 'location'}
        obs, rewards, done, infos = env.step(actions)
        for i, agent in enumerate(agents):
            if i in last_obs:
                out = agent.collect_data(obs.get(i), rewards.get(i, 0), done, i)
                if out is not None:
                    trajs.append(out)
        last_obs = {**last_obs, **obs}
        steps += 1
    assert trajs, 'no trajectory produced'
    assert len(trajs[0]) == 4, 'traj_len steps + bootstrap frame'

    # feed through the learner collate + forward + loss
    batch = rl_collate([trajs[0]])
    batch.pop('model_last_iter', None)
    batch.pop('aux_type', None)
    model = Model(Config({'common': {'type': 'train'},
                          'model': {'enable_baselines':
                                    ['winloss', 'build_order', 'built_unit',
                                     'battle']}}),
                  use_value_network=True)
    out = model.rl_learner_forward(**batch)
    loss = ReinforcementLoss(Config({}), 'MP0').compute_loss(out)
    assert torch.isfinite(loss['total_loss'])
    loss['total_loss'].backward()


@pytest.mark.timeout(600)
def test_actor_standalone_rollout():
    torch.manual_seed(0)
    cfg = Config({'actor': {'episode_num': 1, 'traj_len': 4, 'env_type': 'mock'},
                  'env': {'player_num': 2, 'max_episode_steps': 6},
                  'common': {'experiment_name': 'test_actor', 'type': 'train'}})
    actor = Actor(cfg)
    results = actor.run()
    assert len(results) == 1
    r = results[0]
    assert '0' in r and '1' in r
    assert abs(r['0']['winloss']) == 1 and r['0']['winloss'] == -r['1']['winloss']
    assert 'z_type' in r['0']


def test_agent_stat_data():
    cfg = Config({'common': {'type': 'train'}})
    agent = Agent(cfg)
    agent.reset()
    data = agent.get_stat_data()
    assert 'race_id' in data and 'z_type' in data


@pytest.mark.timeout(600)
def test_actor_respects_pipeline_field():
    """Jobs carrying a non-default pipeline load that agent package through
    the actor's own job setup (docs/agent.md contract)."""

    class StubComm:
        def ask_for_job(self, job_type, player_id=None):
            return {'player_ids': ['MP0', 'TPL'],
                    'pipelines': ['default', 'template'],
                    'checkpoint_paths': ['none', 'none'],
                    'teacher_checkpoint_paths': ['none', 'none'],
                    'z_path': ['3map.json', '3map.json'],
                    'send_data_players': [], 'update_players': [],
                    'env_info': {'map_name': 'KingsCove'}}

        def send_result(self, result):
            return {'ok': True}

        def send_data(self, *a, **k):
            pass

        def pull_model(self, *a, **k):
            return None

    cfg = Config({'actor': {'episode_num': 1, 'env_type': 'mock'},
                  'env': {'player_num': 2, 'max_episode_steps': 4},
                  'common': {'experiment_name': 'test_actor_tpl', 'type': 'train'}})
    actor = Actor(cfg)
    actor._comm = StubComm()
    results = actor.run()
    assert len(results) == 1
    from distar_amd.agents.template.agent import Agent as TemplateAgent
    from distar_amd.actor.agent import Agent as DefaultAgent
    assert isinstance(actor._agents[0], DefaultAgent)
    assert isinstance(actor._agents[1], TemplateAgent)


@pytest.mark.timeout(900)
def test_actor_multi_env_workers():
    """env_num > 1: parallel environment workers sharing the job's models
    (reference forks env_num processes, actor.py:301-319)."""
    torch.manual_seed(0)
    cfg = Config({'actor': {'episode_num': 3, 'env_num': 2, 'env_type': 'mock',
                            'traj_len': 4},
                  'env': {'player_num': 2, 'max_episode_steps': 4},
                  'common': {'experiment_name': 'test_actor_multi',
                             'type': 'train'}})
    actor = Actor(cfg)
    results = actor.run()
    assert len(results) >= 3
    for r in results:
        assert '0' in r and '1' in r


def test_stat_unit_num_tracking():
    """Build/train successes accumulate per-unit-name counts (reference
    lib/stat.py:8-12,47-52); normalized units/* appear in stat data."""
    from distar_amd.lib.actions import ACTIONS, NUM_ACTIONS
    from distar_amd.lib.stat import Stat
    stat = Stat('zerg')
    train_idx = next(i for i in range(NUM_ACTIONS)
                     if ACTIONS[i]['name'].startswith('Train_')
                     and ACTIONS[i]['goal'] in ('build', 'unit'))
    name = ACTIONS[train_idx]['name'].split('_')[1]
    for _ in range(3):
        stat.update(train_idx, 1)       # success
    stat.update(train_idx, 2)           # failure: not counted
    assert stat.unit_num[name] == 3
    assert stat.unit_num['max_unit_num'] == 3
    data = stat.get_stat_data()
    assert data[f'units/{name}'] == 1.0


@pytest.mark.timeout(600)
def test_agent_dapo_successive_logits_flow():
    """With DAPO on and a successive model attached, every rollout step
    carries successive_logit and the RL loss consumes it (reference
    agent.py:506-515,566 + rl_loss.py dapo term)."""
    from distar_amd.losses.rl_loss import ReinforcementLoss
    torch.manual_seed(0)
    cfg = Config({'common': {'type': 'train'},
                  'actor': {'traj_len': 2, 'job_type': 'train'},
                  'learner': {'use_dapo': True},
                  'env': {'player_num': 2, 'max_episode_steps': 100000},
                  'agent': {}})
    env = MockSC2Env(cfg, entity_num_range=(24, 40), seed=3)
    agent = Agent(cfg, env_id=0)
    agent.player_id = 'MP0'
    agent.successive_model = Model(cfg)
    agent.successive_model.eval()
    obs = env.reset()
    agent.reset(obs=obs.get(0))
    trajs, last_obs, done, steps = [], obs, False, 0
    while not done and steps < 6 and not trajs:
        actions = {0: agent.step(last_obs[0])[0]}
        obs, rewards, done, infos = env.step(actions)
        out = agent.collect_data(obs.get(0), rewards.get(0, 0), done, 0)
        if out is not None:
            trajs.append(out)
        last_obs = {**last_obs, **obs}
        steps += 1
    assert trajs
    assert all('successive_logit' in td for td in trajs[0][:-1])
    batch = rl_collate([trajs[0]])
    batch.pop('model_last_iter', None)
    batch.pop('aux_type', None)
    model = Model(Config({'common': {'type': 'train'},
                          'model': {'enable_baselines': ['winloss']}}),
                  use_value_network=True)
    with torch.no_grad():
        out = model.rl_learner_forward(**batch)
        out['successive_logit'] = batch['successive_logit']
        loss = ReinforcementLoss(Config({'learner': {'use_dapo': True},
                                         'model': {'enable_baselines': ['winloss']}}).learner,
                                 'MP0')
        ld = loss.compute_loss(out)
    assert torch.isfinite(ld['total_loss'])
    assert 'dapo/total' in ld or any('dapo' in k for k in ld)


def test_episode_end_window_is_full_length():
    """Episode-end trajectories slide back to a full traj_len window
    (reference agent.py:173 deque(maxlen) semantics) so the learner collate
    always sees uniform T."""
    torch.manual_seed(0)
    cfg = Config({'common': {'type': 'train'},
                  'actor': {'traj_len': 4, 'job_type': 'train'},
                  'env': {'player_num': 1, 'max_episode_steps': 2000},
                  'agent': {}})
    env = MockSC2Env(cfg, entity_num_range=(24, 32), seed=5)
    agent = Agent(cfg, env_id=0)
    agent.player_id = 'MP0'
    obs = env.reset()
    agent.reset(obs=obs.get(0))
    lengths, last_obs, done = [], obs, False
    while not done:
        actions = {0: agent.step(last_obs[0])[0]}
        obs, rewards, done, infos = env.step(actions)
        out = agent.collect_data(obs.get(0), rewards.get(0, 0), done, 0)
        if out is not None:
            lengths.append(len(out))
        last_obs = {**last_obs, **obs}
    assert len(lengths) >= 2, f'need multiple windows, got {lengths}'
    # every emitted window has exactly traj_len steps + bootstrap frame,
    # including the episode-end one (slides back over earlier steps)
    assert all(l == 5 for l in lengths), lengths


def test_battle_pseudo_reward_deltas():
    """Battle reward = delta(own score) - delta(opponent score), /battle_norm
    (reference agent.py:623-626)."""
    import dummy_obs as D
    torch.manual_seed(0)
    cfg = Config({'common': {'type': 'train'}, 'actor': {'traj_len': 4},
                  'env': {'player_num': 2}, 'agent': {}})
    agent = Agent(cfg, env_id=0)
    agent.player_id = 'MP0'
    agent.reset()
    agent._game_info['battle_score'] = 100.
    agent._game_info['opponent_battle_score'] = 50.
    nxt = {'raw_obs': D.raw_observation([D.unit(tag=1)]),
           'opponent_obs': D.raw_observation([D.unit(tag=2)], player_id=2),
           'action_result': [1]}
    from distar_amd.lib.features import compute_battle_score
    own = compute_battle_score(nxt['raw_obs'])
    opp = compute_battle_score(nxt['opponent_obs'])
    _, _, battle = agent._update_fake_reward(0, torch.tensor(0), nxt)
    expected = ((own - 100.) - (opp - 50.)) / 30.
    assert abs(float(battle) - expected) < 1e-6


def test_cum_reward_observation_mode():
    """cum_type='observation': completed own units/upgrades scan into the
    behaviour cumulative stat (reference agent.py:663-677)."""
    import dummy_obs as D
    from distar_amd.lib.actions import UNIT_TO_CUM
    torch.manual_seed(0)
    cfg = Config({'common': {'type': 'train'},
                  'actor': {'traj_len': 4},
                  'agent': {'cum_type': 'observation'},
                  'env': {'player_num': 2}})
    agent = Agent(cfg, env_id=0)
    agent.player_id = 'MP0'
    agent.reset()
    agent.use_cum_reward = True
    # pick a unit type with a cumulative slot
    ut = next(u for u in list(UNIT_TO_CUM.keys()) if UNIT_TO_CUM[u] != -1)
    nxt = {'raw_obs': D.raw_observation(
        [D.unit(tag=1, unit_type=ut, alliance=1)]),
        'action_result': [1]}
    _, cum_reward, _ = agent._update_fake_reward(0, torch.tensor(0), nxt)
    assert agent._behaviour_cumulative_stat[UNIT_TO_CUM[ut]] == 1


def test_cum_reward_cancel_refund():
    """Cancel_quick refunds the cancelled unit's cumulative slot
    (reference agent.py:682-696)."""
    from distar_amd.lib.actions import (ACTIONS, CUMULATIVE_STAT_ACTIONS,
                                        UNIT_ABILITY_TO_ACTION)
    torch.manual_seed(0)
    cfg = Config({'common': {'type': 'train'}, 'actor': {'traj_len': 4},
                  'env': {'player_num': 2}, 'agent': {}})
    agent = Agent(cfg, env_id=0)
    agent.player_id = 'MP0'
    agent.reset()
    agent.use_cum_reward = True
    cancel_at = next(i for i, a in enumerate(ACTIONS)
                     if a['name'] == 'Cancel_quick')
    # choose an order ability that maps to a cumulative action
    ability_idx, mapped = next(
        (k, v) for k, v in UNIT_ABILITY_TO_ACTION.items()
        if v in CUMULATIVE_STAT_ACTIONS)
    ci = CUMULATIVE_STAT_ACTIONS.index(mapped)
    agent._behaviour_cumulative_stat[ci] = 1
    agent._output = {'action_info': {'selected_units': torch.tensor([0])}}
    agent._observation = {'entity_info': {
        'order_length': torch.tensor([1]),
        'order_id_0': torch.tensor([ability_idx])}}
    nxt = {'action_result': [1]}
    agent._update_fake_reward(cancel_at, torch.tensor(0), nxt)
    assert agent._behaviour_cumulative_stat[ci] == 0


def test_agent_loads_real_z_assets():
    """_load_z samples a real strategy-statistics entry from the shipped
    asset files and arms the targets/flags (reference agent.py:176-243)."""
    import random
    torch.manual_seed(0)
    random.seed(0)
    cfg = Config({'common': {'type': 'train'},
                  'actor': {'traj_len': 4},
                  'agent': {'z_path': '3map.json', 'fake_reward_prob': 1.0},
                  'env': {'player_num': 2}})
    agent = Agent(cfg, env_id=0)
    agent.player_id = 'MP0'
    agent.reset(map_name='KingsCove', race='zerg', opponent_race='zerg')
    assert agent._target_building_order.shape[0] >= 1
    assert agent._target_cumulative_stat.shape[0] == 167
    assert int(agent._target_cumulative_stat.sum()) > 0     # real Z content
    assert agent._target_z_loop > 0
    assert agent.use_bo_reward or agent.use_cum_reward
    # style files with z_type annotations load too
    cfg2 = Config({'common': {'type': 'train'}, 'actor': {'traj_len': 4},
                   'agent': {'z_path': 'mutalisk.json'},
                   'env': {'player_num': 2}})
    agent2 = Agent(cfg2, env_id=0)
    agent2.player_id = 'MP0'
    agent2.reset(map_name='KingsCove', race='zerg', opponent_race='zerg')
    assert agent2._target_building_order.shape[0] >= 1
