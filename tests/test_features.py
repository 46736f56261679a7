"""Features transform tests on synthetic protobuf-shaped observations
(reference test strategy: pysc2 dummy_observation + features_test)."""
import torch

import dummy_obs as D
from distar_amd.lib.consts import (ENTITY_INFO, EFFECT_LEN, MAX_ENTITY_NUM,
                                   SCALAR_INFO, SPATIAL_SIZE)
from distar_amd.lib.features import Features, compute_battle_score
from distar_amd.utils.config import Config


def make_features():
    gi = D.game_info()
    units = [D.unit(tag=100 + i, unit_type=86 if i == 0 else 105,
                    alliance=1 if i < 3 else 4, x=20 + i, y=20 + i,
                    orders=(1216,) if i == 0 else ())
             for i in range(6)]
    raw_ob = D.raw_observation(units)
    feat = Features(gi, raw_ob, Config({}))
    return feat, raw_ob, units


def test_transform_obs_schema_complete():
    feat, raw_ob, units = make_features()
    out = feat.transform_obs(raw_ob, padding_spatial=True)
    assert int(out['entity_num']) == 6
    for k, dtype in ENTITY_INFO:
        if 'last' in k:
            continue
        assert k in out['entity_info'], k
        assert out['entity_info'][k].shape[0] == 6
    for k, dtype, size in SCALAR_INFO:
        if k in ('cumulative_stat', 'beginning_order', 'bo_location',
                 'last_queued', 'last_delay', 'last_action_type'):
            continue        # injected by the agent, not the transform
        assert k in out['scalar_info'], k
    for k in ('height_map', 'visibility_map', 'creep', 'player_relative'):
        assert tuple(out['spatial_info'][k].shape) == tuple(SPATIAL_SIZE)
    for k in out['spatial_info']:
        if 'effect' in k:
            assert out['spatial_info'][k].shape[0] == EFFECT_LEN
    assert out['game_info']['tags'] == [100 + i for i in range(6)]
    # y flipped: entity y = map_y - raw y
    assert int(out['entity_info']['y'][0]) == 152 - 20
    # unit type reordered to dense index (<260)
    assert (out['entity_info']['unit_type'] < 260).all()
    # alliance masks drive the scalar bows
    assert int(out['scalar_info']['unit_counts_bow'].sum()) == 3
    assert int(out['scalar_info']['enemy_unit_type_bool'].sum()) >= 1


def test_transform_obs_feeds_model():
    """Transformed obs (after agent-side scalar injection) runs the model."""
    from distar_amd.actor.agent import Agent
    feat, raw_ob, _ = make_features()
    agent = Agent(Config({'common': {'type': 'train'}}))
    agent.reset()
    obs = {'raw_obs': raw_ob, 'game_info_proto': D.game_info(),
           'action_result': [1]}
    action = agent.step(obs)
    assert isinstance(action, list) and 'func_id' in action[0]


def test_reverse_raw_action_roundtrip():
    feat, raw_ob, units = make_features()
    tags = [u.tag for u in units]
    # ability 1216 = general Morph_Hatchery-ish raw_cmd_pt family member;
    # use a known general: 3674 Attack (raw_cmd_pt via Attack_pt func 2)
    act = D.raw_action(ability_id=3674, unit_tags=[100, 101],
                       target_pos=(30.0, 40.0))
    (action_ret, action_mask, su_num, lsu, ltu, invalid) = \
        feat.reverse_raw_action(act, tags)
    assert not invalid
    assert bool(action_mask['action_type'])
    from distar_amd.lib.actions import ACTIONS
    at = int(action_ret['action_type'])
    assert ACTIONS[at]['func_id'] == 2          # Attack_pt
    assert action_ret['selected_units'].tolist() == [0, 1, len(tags)]
    assert int(su_num) == 3
    loc = int(action_ret['target_location'])
    assert loc % SPATIAL_SIZE[1] == 30
    assert loc // SPATIAL_SIZE[1] == 152 - 40


def test_get_z_extraction():
    feat, raw_ob, _ = make_features()
    from distar_amd.lib.actions import BEGINNING_ORDER_ACTIONS, CUMULATIVE_STAT_ACTIONS
    at = BEGINNING_ORDER_ACTIONS[3]
    traj = [{'action_info': {'action_type': torch.tensor(at),
                             'target_location': torch.tensor(500)}}]
    bo, cum, bo_len, bo_loc = feat.get_z(traj)
    assert bo_len == 1 and int(bo[0]) == 3 and int(bo_loc[0]) == 500
    if at in CUMULATIVE_STAT_ACTIONS:
        assert cum[CUMULATIVE_STAT_ACTIONS.index(at)] == 1


def test_battle_score():
    _, raw_ob, _ = make_features()
    # killed minerals 150 + 1.5 * killed vespene 150 (same dummy category sums)
    assert compute_battle_score(raw_ob) == 150. + 1.5 * 150.


def test_transform_obs_value_features():
    """Opponent-side value features (reference features.py:735-765) flow into
    the critic encoder."""
    feat, raw_ob, units = make_features()
    opp_units = [D.unit(tag=900 + i, unit_type=105, alliance=1, x=100 + i, y=80)
                 for i in range(4)]
    opp_ob = D.raw_observation(opp_units, player_id=2)
    out = feat.transform_obs(raw_ob, padding_spatial=True, opponent_obs=opp_ob)
    vf = out['value_feature']
    assert int(vf['total_unit_count']) == 4 + 3     # enemy units + own units
    assert vf['unit_x'].shape[0] == 512
    assert vf['own_units_spatial'].shape == (1, 152, 160)
    assert int(vf['enemy_unit_counts_bow'].sum()) == 4
    # feeds the ValueEncoder (with the behavior-Z keys the agent appends)
    import torch
    from distar_amd.models.alphastar.encoders import ValueEncoder
    from distar_amd.models.alphastar.model import alphastar_model_default_config
    from distar_amd.utils.data import default_collate_with_dim
    enc = ValueEncoder(alphastar_model_default_config)
    vf = dict(vf)
    vf.update({'beginning_order': torch.zeros(20, dtype=torch.long),
               'bo_location': torch.zeros(20, dtype=torch.long),
               'cumulative_stat': torch.zeros(167, dtype=torch.long)})
    batch = default_collate_with_dim([vf, vf])
    out_v = enc(batch)
    assert out_v.shape == (2, 544)


def test_transform_obs_passengers_become_entities():
    """Units riding in transports appear as in-cargo entity rows (reference
    features.py:544-560): carried units share the carrier's position/owner and
    set the is_in_cargo field."""
    gi = D.game_info()
    carrier = D.unit(tag=200, unit_type=86, x=40, y=40,
                     passengers=[D.passenger(tag=300), D.passenger(tag=301)])
    raw_ob = D.raw_observation([carrier])
    feat = Features(gi, raw_ob, Config({}))
    out = feat.transform_obs(raw_ob, padding_spatial=True)
    n = int(out['entity_num'])
    assert n == 3                                   # carrier + 2 passengers
    cargo = out['entity_info']['is_in_cargo'][:n]
    assert cargo.tolist() == [0, 1, 1]
    # passengers inherit the carrier's location
    assert out['entity_info']['x'][1] == out['entity_info']['x'][0]
    assert out['entity_info']['y'][2] == out['entity_info']['y'][0]


def test_transform_obs_effects_encode_locations():
    """Active effects land in the effect_* spatial rasters (reference
    features.py:479-487), except own liberator/lurker zones."""
    feat, _, _ = make_features()
    psistorm, liberator = 1, 9
    raw_ob = D.raw_observation(
        [D.unit(tag=100)],
        effects=[(psistorm, 2, 50, 60), (liberator, 1, 30, 30),
                 (liberator, 2, 70, 80)])
    out = feat.transform_obs(raw_ob, padding_spatial=True)
    storm = out['spatial_info']['effect_PsiStorm']
    assert storm.shape == (EFFECT_LEN,)
    loc = int(50) + int(feat.map_size.y - 60) * 160
    assert loc in storm.tolist()
    # own (owner==1) LiberatorDefenderZone is filtered, enemy kept
    lib = out['spatial_info']['effect_LiberatorDefenderZone'].tolist()
    own_loc = 30 + int(feat.map_size.y - 30) * 160
    enemy_loc = 70 + int(feat.map_size.y - 80) * 160
    assert enemy_loc in lib and own_loc not in lib


def test_transform_obs_upgrades_one_hot():
    """Researched upgrade ids scatter into the NUM_UPGRADES one-hot
    (reference features.py:505-512)."""
    feat, _, _ = make_features()
    raw_ob = D.raw_observation([D.unit(tag=100)], upgrades=(1, 3))
    out = feat.transform_obs(raw_ob, padding_spatial=True)
    up = out['scalar_info']['upgrades']
    assert int(up.sum()) == 2
    assert up[1] == 1 and up[3] == 1


def test_filter_actions_dedups_spam():
    """Spammed identical train/morph/research commands inside the window
    collapse to the LAST occurrence; distinct tags/abilities and spaced
    repeats survive (reference replay_decoder.py:70-213)."""
    from distar_amd.data.replay_decoder import FilterActions
    from distar_amd.lib.actions import ACTIONS
    f = FilterActions()
    train_gab = next(a['general_ability_id'] for a in ACTIONS
                     if a['goal'] == 'unit' and a['general_ability_id'])
    other_gab = next(a['general_ability_id'] for a in ACTIONS
                     if a['goal'] == 'build' and a['general_ability_id']
                     and 'Morph' not in a['name'])
    tags = (10, 11)
    seq = [
        (0, train_gab, tags, 'a0'),
        (1, train_gab, tags, 'a1'),      # spam: within window, same tags
        (2, train_gab, tags, 'a2'),      # spam again -> keep only a2
        (3, train_gab, (99,), 'b0'),     # different tags: kept
        (4, other_gab, tags, 'c0'),      # non-filtered goal: kept
        (20, train_gab, tags, 'd0'),     # far outside window: kept
    ]
    out = f.run(seq)
    kept = [x[3] for x in out]
    assert kept == ['a2', 'b0', 'c0', 'd0']
    # the surviving spam entry carries the LAST loop (freshest delay timing)
    assert out[0][0] == 2


def test_reverse_raw_action_families():
    """Target-unit, quick (no-target), camera-move and special remaps
    (unload->3664, cancel-slot->3671, frivolous dropped) decode per the
    reference (features.py:853-952)."""
    from distar_amd.lib.actions import ACTIONS
    feat, raw_ob, units = make_features()
    tags = [u.tag for u in units]

    # raw_cmd_unit: attack a specific unit
    act = D.raw_action(ability_id=3674, unit_tags=[100], target_unit_tag=102)
    action_ret, mask, su_num, lsu, ltu, invalid = feat.reverse_raw_action(act, tags)
    assert not invalid
    assert int(action_ret['target_unit']) == tags.index(102)
    assert ACTIONS[int(action_ret['action_type'])]['func_id'] == 3   # Attack_unit

    # raw_cmd (quick): stop
    act = D.raw_action(ability_id=3665, unit_tags=[100])
    action_ret, *_, invalid = feat.reverse_raw_action(act, tags)
    assert not invalid and action_ret['action_type'] is not None
    assert ACTIONS[int(action_ret['action_type'])]['name'].startswith('Stop')

    # frivolous abilities (6/7) are dropped -> masked default + invalid
    act = D.raw_action(ability_id=6, unit_tags=[100])
    action_ret, mask, _, _, _, invalid = feat.reverse_raw_action(act, tags)
    assert invalid and not bool(mask['action_type'])

    # target-unit tag missing from the obs -> invalid flagged
    act = D.raw_action(ability_id=3674, unit_tags=[100], target_unit_tag=999999)
    *_, invalid = feat.reverse_raw_action(act, tags)
    assert invalid


def test_agent_emits_world_coordinates():
    """_post_process inverts the y axis back to game-world coordinates
    (reference agent.py:389-391): transform_obs in / action out roundtrip."""
    import dummy_obs as D
    from distar_amd.actor.agent import Agent
    from distar_amd.lib.actions import ACTIONS
    torch.manual_seed(0)
    cfg = Config({'common': {'type': 'train'}, 'actor': {'traj_len': 4},
                  'env': {'player_num': 2}, 'agent': {}})
    agent = Agent(cfg, env_id=0)
    agent.player_id = 'MP0'
    gi = D.game_info()
    # exactly one own hatchery (type 86) = the born base; drones otherwise
    units = [D.unit(tag=100, unit_type=86)] + \
        [D.unit(tag=101 + i, unit_type=104) for i in range(2)]
    raw_ob = D.raw_observation(units)
    obs = {'game_info_proto': gi, 'raw_obs': raw_ob, 'game_info': gi}
    agent.reset(obs=obs)
    action = agent.step(obs)[0]
    x, y_world = action['location']
    # the model's flat location index for this action
    loc = int(agent._output['action_info']['target_location'])
    y_model = loc // 160
    assert y_world == max(agent._feature.map_size.y - y_model, 0)
    assert x == loc % 160


def test_action_table_invariants():
    """Derived action-table structures match the reference derivations
    (reference lib/actions.py:355-426); guards the data assets."""
    from distar_amd.lib.actions import (ACTIONS, NUM_ACTIONS, QUEUE_ACTIONS,
                                        BEGINNING_ORDER_ACTIONS,
                                        CUMULATIVE_STAT_ACTIONS,
                                        SELECTED_UNITS_MASK, UNIT_TO_CUM,
                                        UPGRADE_TO_CUM)
    assert NUM_ACTIONS == 327 and len(ACTIONS) == 327
    assert len(QUEUE_ACTIONS) == 109          # 'Train_'/'Research' actions
    assert len(BEGINNING_ORDER_ACTIONS) == 174
    assert len(CUMULATIVE_STAT_ACTIONS) == 167
    assert int(SELECTED_UNITS_MASK.sum()) == 325
    # every queue action really is a train/research action
    assert all('Train_' in ACTIONS[i]['name'] or 'Research' in ACTIONS[i]['name']
               for i in QUEUE_ACTIONS)
    # cum LUTs map into valid slots
    assert all(0 <= v < 167 for v in UNIT_TO_CUM.values() if v != -1)
    assert all(0 <= v < 167 for v in UPGRADE_TO_CUM.values() if v != -1)
    # action schema fields present on every row
    for a in ACTIONS:
        assert {'name', 'func_id', 'general_ability_id', 'goal', 'queued',
                'selected_units', 'target_location', 'target_unit'} <= set(a)
