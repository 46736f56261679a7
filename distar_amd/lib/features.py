"""Observation/action transform between the SC2 protocol and the model's
tensor schemas.

Functional parity with the reference's `distar/agent/default/lib/features.py`
(`Features.transform_obs:462-767`, `reverse_raw_action:853-952`,
`get_z:419-460`, `compute_battle_score:354-363`).  Written duck-typed against
the protobuf attribute structure so it runs on real `s2clientprotocol`
messages AND on synthetic test doubles (`tests/dummy_obs.py`, the pysc2
dummy-observation idea) — no protobuf import needed here.

The raw-function table (`assets/raw_functions.json`, extracted game data)
supplies ability->general->function-type resolution for decoding replay
actions.
"""
import json
import os
import random
from collections import defaultdict

import numpy as np
import torch
from torch import int16, uint8

from .actions import (ACTIONS, ABILITY_TO_QUEUE_ACTION, BEGINNING_ORDER_ACTIONS,
                      CUMULATIVE_STAT_ACTIONS, FUNC_ID_TO_ACTION_TYPE_DICT,
                      NUM_UNIT_MIX_ABILITIES, UNIT_ABILITY_REORDER)
from .consts import (ACTION_INFO, BEGINNING_ORDER_LENGTH, EFFECT_LEN,
                     ENTITY_INFO, MAX_ENTITY_NUM, MAX_SELECTED_UNITS_NUM,
                     SPATIAL_INFO, SPATIAL_SIZE, UPGRADE_LENGTH)
from ..utils.timing import sw
from .static_data import (ADDON_REORDER_ARRAY, BUFFS_REORDER_ARRAY,
                          NUM_UNIT_TYPES, NUM_UPGRADES,
                          UNIT_TYPES_REORDER_ARRAY, UPGRADES_REORDER_ARRAY)

# effect_id -> name for the 6 tracked effects (SC2 data enum subset)
EFFECT_ID_TO_NAME = {
    1: 'PsiStorm', 7: 'NukeDot', 9: 'LiberatorDefenderZone',
    10: 'BlindingCloud', 11: 'CorrosiveBile', 12: 'LurkerSpines',
}
EFFECT_NAMES = {'PsiStorm': 1, 'NukeDot': 7, 'LiberatorDefenderZone': 9,
                'BlindingCloud': 10, 'CorrosiveBile': 11, 'LurkerSpines': 12}

SCORE_CATEGORIES = ['none', 'army', 'economy', 'technology', 'upgrade']

MINIMAP_NAMES = ['height_map', 'visibility_map', 'creep', 'player_relative',
                 'alerts', 'pathable', 'buildable']

_ASSET_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), 'assets')
with open(os.path.join(_ASSET_DIR, 'raw_functions.json')) as _f:
    RAW_FUNCTIONS = json.load(_f)['raw_functions']

# ability -> list of raw functions; generalized ability chase
RAW_ABILITY_IDS = defaultdict(list)
for _func in RAW_FUNCTIONS:
    if _func['ability_id']:
        RAW_ABILITY_IDS[_func['ability_id']].append(_func)

# func_id -> raw function entry (env.transform_action routing)
RAW_FUNC_BY_ID = {f['id']: f for f in RAW_FUNCTIONS}


def unpack_feature_layer(image_data):
    """Decode a protobuf ImageData-like object (bits_per_pixel, size, data)
    to a (y, x) numpy array (pysc2 features.Feature.unpack equivalent)."""
    bpp = image_data.bits_per_pixel
    y, x = image_data.size.y, image_data.size.x
    data = np.frombuffer(image_data.data, dtype=np.uint8)
    if bpp == 1:
        data = np.unpackbits(data)[:y * x]
    elif bpp == 8:
        pass
    elif bpp == 16:
        data = np.frombuffer(image_data.data, dtype=np.uint16)
    elif bpp == 32:
        data = np.frombuffer(image_data.data, dtype=np.int32)
    else:
        raise ValueError(f'unsupported bits_per_pixel {bpp}')
    return data.reshape(y, x).copy()


def compute_battle_score(obs):
    if obs is None:
        return 0.
    sd = obs.observation.score.score_details
    killed_mineral = sum(getattr(sd.killed_minerals, s) for s in SCORE_CATEGORIES)
    killed_vespene = sum(getattr(sd.killed_vespene, s) for s in SCORE_CATEGORIES)
    return killed_mineral + 1.5 * killed_vespene


class Features:
    """Per-game transform state (map size, races, born locations)."""

    def __init__(self, game_info, raw_ob=None, cfg=None):
        cfg = cfg or {}
        self._map_size = game_info.start_raw.map_size
        self._requested_races = {
            info.player_id: info.race_requested for info in game_info.player_info
            if getattr(info, 'type', 1) != 3}          # 3 = Observer
        self._map_name = game_info.map_name
        start_locations = list(game_info.start_raw.start_locations)
        self._start_location = start_locations[0] if start_locations else None
        self._whole_cfg = cfg
        self._cfg = cfg.get('feature', {}) if hasattr(cfg, 'get') else {}
        self._bo_zergling_num = self._cfg.get('bo_zergling_num', 8)
        self._beginning_order_flag = random.random() < self._cfg.get('beginning_order_prob', 1.)
        self._cumulative_stat_flag = random.random() < self._cfg.get('cumulative_stat_prob', 1.)
        self._zero_z_value = self._cfg.get('zero_z_value', 1.)
        self._filter_spine = self._cfg.get('filter_spine', True)
        if raw_ob is not None:
            self._init_born_location(game_info, raw_ob)
        else:
            self._born_location = 0
            self._away_born_location = 0

    def _init_born_location(self, game_info, raw_ob):
        location = [[u.pos.x, u.pos.y] for u in raw_ob.observation.raw_data.units
                    if u.unit_type in (59, 18, 86)]
        assert len(location) == 1, 'corrupt replay: no fog of war'
        born = location[0]
        self._born_location = int(born[0]) + \
            int(self.map_size.y - born[1]) * SPATIAL_SIZE[1]
        away = game_info.start_raw.start_locations[0]
        self._away_born_location = int(away.x) + \
            int(self.map_size.y - away.y) * SPATIAL_SIZE[1]

    map_size = property(lambda self: self._map_size)
    map_name = property(lambda self: self._map_name)
    requested_races = property(lambda self: self._requested_races)
    home_born_location = property(lambda self: self._born_location)
    away_born_location = property(lambda self: self._away_born_location)

    # ------------------------------------------------------------------- obs
    @sw.decorate('transform_obs')
    def transform_obs(self, obs, padding_spatial=False, opponent_obs=None):
        spatial_info = defaultdict(list)
        scalar_info = {}
        entity_info = {}
        game_info = {}
        raw = obs.observation.raw_data

        # spatial planes
        mm = obs.observation.feature_layer_data.minimap_renders
        for name in MINIMAP_NAMES:
            d = torch.from_numpy(unpack_feature_layer(getattr(mm, name)))
            pad_y = SPATIAL_SIZE[0] - d.shape[0]
            pad_x = SPATIAL_SIZE[1] - d.shape[1]
            if (pad_y or pad_x) and padding_spatial:
                d = torch.nn.functional.pad(d, (0, pad_x, 0, pad_y), 'constant', 0)
            spatial_info[name] = d
        for e in raw.effects:
            name = EFFECT_ID_TO_NAME.get(e.effect_id)
            if name is None:
                continue
            if name in ('LiberatorDefenderZone', 'LurkerSpines') and e.owner == 1:
                continue
            for p in e.pos:
                loc = int(p.x) + int(self.map_size.y - p.y) * SPATIAL_SIZE[1]
                spatial_info['effect_' + name].append(loc)
        for k, _ in SPATIAL_INFO:
            if 'effect' in k:
                vals = spatial_info[k][:EFFECT_LEN]
                vals += [0] * (EFFECT_LEN - len(vals))
                spatial_info[k] = torch.as_tensor(vals, dtype=int16)

        # entity rows
        tag_types = {}

        def get_addon_type(tag):
            if not tag_types:
                for u in raw.units:
                    tag_types[u.tag] = u.unit_type
            return tag_types.get(tag, 0)

        tags, units = [], []
        for u in raw.units:
            tags.append(u.tag)
            orders = list(u.orders)
            units.append([
                u.unit_type, u.alliance, u.cargo_space_taken, u.build_progress,
                u.health_max, u.shield_max, u.energy_max, u.display_type,
                u.owner, u.pos.x, u.pos.y, u.cloak, u.is_blip, u.is_powered,
                u.mineral_contents, u.vespene_contents, u.cargo_space_max,
                u.assigned_harvesters, u.weapon_cooldown, len(orders),
                orders[0].ability_id if len(orders) > 0 else 0,
                orders[1].ability_id if len(orders) > 1 else 0,
                u.is_hallucination,
                u.buff_ids[0] if len(u.buff_ids) >= 1 else 0,
                u.buff_ids[1] if len(u.buff_ids) >= 2 else 0,
                get_addon_type(u.add_on_tag) if u.add_on_tag else 0,
                u.is_active,
                orders[0].progress if len(orders) >= 1 else 0,
                orders[1].progress if len(orders) >= 2 else 0,
                orders[2].ability_id if len(orders) > 2 else 0,
                orders[3].ability_id if len(orders) > 3 else 0,
                0,  # is_in_cargo
                u.attack_upgrade_level, u.armor_upgrade_level,
                u.shield_upgrade_level, u.health, u.shield, u.energy,
            ])
            for v in u.passengers:
                tags.append(v.tag)
                units.append([
                    v.unit_type, u.alliance, 0, 0, v.health_max, v.shield_max,
                    v.energy_max, 0, u.owner, u.pos.x, u.pos.y, 0, 0, 0, 0, 0,
                    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
                    1, 0, 0, 0, v.health, v.shield, v.energy,
                ])
        units = units[:MAX_ENTITY_NUM]
        tags = tags[:MAX_ENTITY_NUM]
        cols = ['unit_type', 'alliance', 'cargo_space_taken', 'build_progress',
                'health_max', 'shield_max', 'energy_max', 'display_type',
                'owner', 'x', 'y', 'cloak', 'is_blip', 'is_powered',
                'mineral_contents', 'vespene_contents', 'cargo_space_max',
                'assigned_harvesters', 'weapon_cooldown', 'order_length',
                'order_id_0', 'order_id_1', 'is_hallucination', 'buff_id_0',
                'buff_id_1', 'addon_unit_type', 'is_active', 'order_progress_0',
                'order_progress_1', 'order_id_2', 'order_id_3', 'is_in_cargo',
                'attack_upgrade_level', 'armor_upgrade_level',
                'shield_upgrade_level', 'health', 'shield', 'energy']
        arr = np.asarray(units, dtype=np.float32).reshape(-1, len(cols))
        col = {name: torch.from_numpy(arr[:, i].copy()) for i, name in enumerate(cols)}

        for k, dtype in ENTITY_INFO:
            if 'last' in k:
                continue
            if k == 'unit_type':
                entity_info[k] = UNIT_TYPES_REORDER_ARRAY[col[k].long()].short()
            elif k == 'order_id_0':
                entity_info[k] = UNIT_ABILITY_REORDER[col[k].long()].short()
            elif k in ('order_id_1', 'order_id_2', 'order_id_3'):
                entity_info[k] = ABILITY_TO_QUEUE_ACTION[col[k].long()].short()
            elif 'buff_id' in k:
                entity_info[k] = BUFFS_REORDER_ARRAY[col[k].long()].short()
            elif k == 'addon_unit_type':
                entity_info[k] = ADDON_REORDER_ARRAY[col[k].long()].short()
            elif k in ('cargo_space_taken', 'cargo_space_max'):
                entity_info[k] = col[k].to(dtype).clamp_(0, 8)
            elif k == 'health_ratio':
                entity_info[k] = (col['health'] / (col['health_max'] + 1e-6)).to(dtype)
            elif k == 'shield_ratio':
                entity_info[k] = (col['shield'] / (col['shield_max'] + 1e-6)).to(dtype)
            elif k == 'energy_ratio':
                entity_info[k] = (col['energy'] / (col['energy_max'] + 1e-6)).to(dtype)
            elif k == 'mineral_contents':
                entity_info[k] = (col[k] / 1800).to(dtype)
            elif k == 'vespene_contents':
                entity_info[k] = (col[k] / 2500).to(dtype)
            elif k == 'y':
                entity_info[k] = (self.map_size.y - col['y']).to(dtype)
            else:
                entity_info[k] = col[k].to(dtype)

        # scalar features
        scalar_info['time'] = torch.tensor(obs.observation.game_loop, dtype=torch.float)
        player = obs.observation.player_common
        stats = torch.tensor([
            player.minerals, player.vespene, player.food_used, player.food_cap,
            player.food_army, player.food_workers, player.idle_worker_count,
            player.army_count, player.warp_gate_count, player.larva_count],
            dtype=torch.float)
        scalar_info['agent_statistics'] = torch.log(stats + 1)
        scalar_info['home_race'] = torch.tensor(
            self._requested_races[player.player_id], dtype=uint8)
        for pid, race in self._requested_races.items():
            if pid != player.player_id:
                scalar_info['away_race'] = torch.tensor(race, dtype=uint8)
        upgrades = torch.zeros(NUM_UPGRADES, dtype=uint8)
        raw_upgrades = UPGRADES_REORDER_ARRAY[
            torch.as_tensor(list(raw.player.upgrade_ids)[:UPGRADE_LENGTH],
                            dtype=torch.long)]
        upgrades.scatter_(0, raw_upgrades, 1)
        scalar_info['upgrades'] = upgrades
        own_types = entity_info['unit_type'][entity_info['alliance'] == 1]
        bow = torch.zeros(NUM_UNIT_TYPES, dtype=uint8)
        scalar_info['unit_counts_bow'] = torch.scatter_add(
            bow, 0, own_types.long(), torch.ones_like(own_types, dtype=uint8))
        scalar_info['unit_type_bool'] = (scalar_info['unit_counts_bow'] > 0).to(uint8)
        scalar_info['unit_order_type'] = torch.zeros(NUM_UNIT_MIX_ABILITIES, dtype=uint8)
        own_orders = entity_info['order_id_0'][entity_info['alliance'] == 1]
        scalar_info['unit_order_type'].scatter_(
            0, own_orders.long().clamp(min=0), torch.ones_like(own_orders, dtype=uint8))
        enemy_types = entity_info['unit_type'][entity_info['alliance'] == 4]
        scalar_info['enemy_unit_type_bool'] = torch.scatter(
            torch.zeros(NUM_UNIT_TYPES, dtype=uint8), 0, enemy_types.long(),
            torch.ones_like(enemy_types, dtype=uint8))

        game_info['map_name'] = self._map_name
        game_info['action_result'] = [o.result for o in obs.action_errors] or [1]
        game_info['game_loop'] = obs.observation.game_loop
        game_info['tags'] = tags
        game_info['battle_score'] = compute_battle_score(obs)
        game_info['opponent_battle_score'] = 0.
        ret = {'spatial_info': dict(spatial_info), 'scalar_info': scalar_info,
               'entity_num': torch.tensor(len(units), dtype=torch.long),
               'entity_info': entity_info, 'game_info': game_info}

        if opponent_obs is not None:
            ret['value_feature'] = self._value_feature(opponent_obs, entity_info,
                                                       own_types, ret)
            game_info['opponent_battle_score'] = compute_battle_score(opponent_obs)
        return ret

    def _value_feature(self, opponent_obs, entity_info, own_types, ret):
        raw = opponent_obs.observation.raw_data
        enemy_x, enemy_y, enemy_types, alliance = [], [], [], []
        for u in raw.units:
            if u.alliance == 1:
                enemy_x.append(u.pos.x)
                enemy_y.append(u.pos.y)
                enemy_types.append(u.unit_type)
                alliance.append(1)
        enemy_types = UNIT_TYPES_REORDER_ARRAY[
            torch.as_tensor(enemy_types, dtype=torch.long)].short()
        bow = torch.zeros(NUM_UNIT_TYPES, dtype=uint8)
        enemy_bow = torch.scatter_add(bow, 0, enemy_types.long(),
                                      torch.ones_like(enemy_types, dtype=uint8))
        own_mask = entity_info['alliance'] == 1
        unit_type = torch.cat([enemy_types, own_types])
        unit_x = torch.cat([torch.as_tensor(enemy_x, dtype=uint8),
                            entity_info['x'][own_mask]])
        enemy_y = (self.map_size.y - torch.as_tensor(enemy_y, dtype=torch.float)).to(uint8)
        unit_y = torch.cat([enemy_y, entity_info['y'][own_mask]])
        total = len(unit_y)
        alliance += [0] * (total - len(alliance))
        alliance = torch.as_tensor(alliance, dtype=torch.bool)
        pad = MAX_ENTITY_NUM - total
        if pad > 0:
            unit_x = torch.nn.functional.pad(unit_x, (0, pad))
            unit_y = torch.nn.functional.pad(unit_y, (0, pad))
            unit_type = torch.nn.functional.pad(unit_type, (0, pad))
            alliance = torch.nn.functional.pad(alliance, (0, pad))
        else:
            unit_x, unit_y = unit_x[:MAX_ENTITY_NUM], unit_y[:MAX_ENTITY_NUM]
            unit_type, alliance = unit_type[:MAX_ENTITY_NUM], alliance[:MAX_ENTITY_NUM]
        player = opponent_obs.observation.player_common
        stats = torch.tensor([
            player.minerals, player.vespene, player.food_used, player.food_cap,
            player.food_army, player.food_workers, player.idle_worker_count,
            player.army_count, player.warp_gate_count, player.larva_count],
            dtype=torch.float)
        enemy_upgrades = torch.zeros(NUM_UPGRADES, dtype=uint8)
        raw_up = UPGRADES_REORDER_ARRAY[
            torch.as_tensor(list(raw.player.upgrade_ids)[:UPGRADE_LENGTH],
                            dtype=torch.long)]
        enemy_upgrades.scatter_(0, raw_up, 1)
        mm = opponent_obs.observation.feature_layer_data.minimap_renders
        d = torch.from_numpy(unpack_feature_layer(mm.player_relative))
        pad_y = SPATIAL_SIZE[0] - d.shape[0]
        pad_x = SPATIAL_SIZE[1] - d.shape[1]
        if pad_y or pad_x:
            d = torch.nn.functional.pad(d, (0, pad_x, 0, pad_y), 'constant', 0)
        enemy_units_spatial = d == 1
        own_units_spatial = ret['spatial_info']['player_relative'] == 1
        return {'unit_type': unit_type, 'enemy_unit_counts_bow': enemy_bow,
                'enemy_unit_type_bool': (enemy_bow > 0).to(uint8),
                'unit_x': unit_x, 'unit_y': unit_y, 'unit_alliance': alliance,
                'total_unit_count': torch.tensor(total, dtype=torch.long),
                'enemy_agent_statistics': torch.log(stats + 1),
                'enemy_upgrades': enemy_upgrades,
                'own_units_spatial': own_units_spatial.unsqueeze(0),
                'enemy_units_spatial': enemy_units_spatial.unsqueeze(0)}

    # --------------------------------------------------------------------- Z
    def get_z(self, traj_data):
        """Build-order + cumulative-stat statistics from a decoded trajectory
        (reference features.py:419-460)."""
        zergling_count = 0
        beginning_order, bo_location = [], []
        cumulative_stat = torch.zeros(len(CUMULATIVE_STAT_ACTIONS), dtype=torch.int8)
        own_x = self.home_born_location % SPATIAL_SIZE[1]
        own_y = self.home_born_location // SPATIAL_SIZE[1]
        away_x = self.away_born_location % SPATIAL_SIZE[1]
        away_y = self.away_born_location // SPATIAL_SIZE[1]
        for step_data in traj_data:
            action_type = int(step_data['action_info']['action_type'])
            if action_type == 322:
                zergling_count += 1
                if zergling_count > self._bo_zergling_num:
                    continue
            if action_type in BEGINNING_ORDER_ACTIONS:
                location = int(step_data['action_info']['target_location'])
                if self._filter_spine and action_type == 54:
                    x, y = location % SPATIAL_SIZE[1], location // SPATIAL_SIZE[1]
                    if (own_x - x) ** 2 + (own_y - y) ** 2 < \
                            (away_x - x) ** 2 + (away_y - y) ** 2:
                        continue
                beginning_order.append(BEGINNING_ORDER_ACTIONS.index(action_type))
                bo_location.append(location)
            if action_type in CUMULATIVE_STAT_ACTIONS:
                cumulative_stat[CUMULATIVE_STAT_ACTIONS.index(action_type)] = 1
        bo_len = len(beginning_order)
        beginning_order = (beginning_order + [0] * BEGINNING_ORDER_LENGTH)[:BEGINNING_ORDER_LENGTH]
        bo_location = (bo_location + [0] * BEGINNING_ORDER_LENGTH)[:BEGINNING_ORDER_LENGTH]
        beginning_order = self._beginning_order_flag * \
            torch.as_tensor(beginning_order, dtype=torch.short)
        bo_location = self._beginning_order_flag * \
            torch.as_tensor(bo_location, dtype=torch.short)
        if not self._cumulative_stat_flag:
            cumulative_stat = 0 * cumulative_stat + self._zero_z_value
        return beginning_order, cumulative_stat, bo_len, bo_location

    # ------------------------------------------------------------- actions
    def reverse_raw_action(self, action, raw_tags):
        """Protobuf action -> 6-head labels (reference features.py:853-952)."""
        action_ret = {'action_type': None,
                      'delay': torch.tensor(0, dtype=torch.long),
                      'queued': None, 'selected_units': None,
                      'target_unit': None, 'target_location': None}
        last_selected_unit_tags = None
        last_target_unit_tag = None
        invalid = False
        units, tags = [], []

        def transfer_action_type(ability_id, cmd_type):
            cancel_slot = {313, 1039, 305, 307, 309, 1832, 1834, 3672}
            unload_unit = {410, 415, 397, 1440, 2373, 1409, 914, 3670}
            frivolous = {6, 7}
            if ability_id in frivolous:
                return None
            if ability_id in unload_unit:
                ability_id = 3664
            elif ability_id in cancel_slot:
                ability_id = 3671
            funcs = RAW_ABILITY_IDS.get(ability_id)
            if not funcs:
                return None
            general = funcs[0]['general_id']
            if general:
                ability_id = general
                funcs = RAW_ABILITY_IDS.get(ability_id, funcs)
            for func in funcs:
                if func['function_type'] == cmd_type:
                    return FUNC_ID_TO_ACTION_TYPE_DICT.get(func['id'])
            return None

        raw_act = action.action_raw
        if raw_act.HasField('unit_command'):
            uc = raw_act.unit_command
            action_ret['queued'] = torch.tensor(uc.queue_command, dtype=torch.long)
            for t in uc.unit_tags:
                if t in raw_tags:
                    units.append(raw_tags.index(t))
                    tags.append(t)
            if uc.HasField('target_unit_tag'):
                if uc.target_unit_tag in raw_tags:
                    action_ret['target_unit'] = torch.tensor(
                        raw_tags.index(uc.target_unit_tag), dtype=torch.long)
                    last_target_unit_tag = uc.target_unit_tag
                else:
                    invalid = True
                action_ret['action_type'] = transfer_action_type(
                    uc.ability_id, 'raw_cmd_unit')
            elif uc.HasField('target_world_space_pos'):
                x = min(int(uc.target_world_space_pos.x), self.map_size.x - 1)
                y = min(self.map_size.y - int(uc.target_world_space_pos.y),
                        self.map_size.y - 1)
                action_ret['target_location'] = torch.tensor(
                    y * SPATIAL_SIZE[1] + x, dtype=torch.long)
                action_ret['action_type'] = transfer_action_type(
                    uc.ability_id, 'raw_cmd_pt')
            else:
                action_ret['action_type'] = transfer_action_type(
                    uc.ability_id, 'raw_cmd')
        if raw_act.HasField('toggle_autocast'):
            ta = raw_act.toggle_autocast
            action_ret['action_type'] = transfer_action_type(
                ta.ability_id, 'raw_autocast')
            for t in ta.unit_tags:
                if t in raw_tags:
                    units.append(raw_tags.index(t))
                    tags.append(t)

        if action_ret['action_type'] is not None:
            action_ret['action_type'] = torch.tensor(
                action_ret['action_type'], dtype=torch.long)
        else:
            invalid = True
        if units and not invalid:
            last_selected_unit_tags = tags
            units.append(len(raw_tags))                  # end flag
            action_ret['selected_units'] = torch.tensor(units, dtype=torch.long)
            selected_units_num = torch.tensor(len(units), dtype=torch.long)
        else:
            invalid = True
            selected_units_num = torch.tensor(0, dtype=torch.long)
        action_mask = {}
        for k, v in action_ret.items():
            if v is None:
                action_mask[k] = torch.tensor(0, dtype=torch.bool)
                action_ret[k] = torch.tensor([0], dtype=torch.long) \
                    if k == 'selected_units' else ACTION_INFO[k]
            else:
                action_mask[k] = torch.tensor(1, dtype=torch.bool)
        action_ret['selected_units'] = \
            action_ret['selected_units'][:MAX_SELECTED_UNITS_NUM]
        selected_units_num.clamp_(max=MAX_SELECTED_UNITS_NUM)
        return (action_ret, action_mask, selected_units_num,
                last_selected_unit_tags, last_target_unit_tag, invalid)
