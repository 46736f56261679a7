"""Observation/action tensor schemas — the data contract every layer compiles
against.  Constants and schemas follow the reference verbatim
(`distar/agent/default/lib/features.py:31-145`); these are wire-format facts,
not code.  ``fake_step_data``/``fake_rl_step_data``/``fake_model_output``
build synthetic batches of exactly that shape for tests, shared-memory slab
allocation and benchmarks.
"""
import copy

import torch
from torch import uint8, int8, int16, float16, float32

from .actions import (NUM_ACTIONS, NUM_CUMULATIVE_STAT_ACTIONS, NUM_UNIT_MIX_ABILITIES)
from .static_data import NUM_UNIT_TYPES, NUM_UPGRADES

SPATIAL_SIZE = [152, 160]  # y, x
BUFF_LENGTH = 3
UPGRADE_LENGTH = 20
MAX_DELAY = 127
BEGINNING_ORDER_LENGTH = 20
MAX_SELECTED_UNITS_NUM = 64
MAX_ENTITY_NUM = 512
EFFECT_LEN = 100
DEFAULT_SPATIAL_SIZE = SPATIAL_SIZE

SPATIAL_INFO = [
    ('height_map', uint8), ('visibility_map', uint8), ('creep', uint8),
    ('player_relative', uint8), ('alerts', uint8), ('pathable', uint8),
    ('buildable', uint8), ('effect_PsiStorm', int16), ('effect_NukeDot', int16),
    ('effect_LiberatorDefenderZone', int16), ('effect_BlindingCloud', int16),
    ('effect_CorrosiveBile', int16), ('effect_LurkerSpines', int16),
]

# (name, dtype, shape)
SCALAR_INFO = [
    ('home_race', uint8, ()), ('away_race', uint8, ()),
    ('upgrades', int16, (NUM_UPGRADES,)), ('time', float32, ()),
    ('unit_counts_bow', uint8, (NUM_UNIT_TYPES,)),
    ('agent_statistics', float32, (10,)),
    ('cumulative_stat', uint8, (NUM_CUMULATIVE_STAT_ACTIONS,)),
    ('beginning_order', int16, (BEGINNING_ORDER_LENGTH,)),
    ('last_queued', int16, ()), ('last_delay', int16, ()),
    ('last_action_type', int16, ()),
    ('bo_location', int16, (BEGINNING_ORDER_LENGTH,)),
    ('unit_order_type', uint8, (NUM_UNIT_MIX_ABILITIES,)),
    ('unit_type_bool', uint8, (NUM_UNIT_TYPES,)),
    ('enemy_unit_type_bool', uint8, (NUM_UNIT_TYPES,)),
]

ENTITY_INFO = [
    ('unit_type', int16), ('alliance', uint8), ('cargo_space_taken', uint8),
    ('build_progress', float16), ('health_ratio', float16), ('shield_ratio', float16),
    ('energy_ratio', float16), ('display_type', uint8), ('x', uint8), ('y', uint8),
    ('cloak', uint8), ('is_blip', uint8), ('is_powered', uint8),
    ('mineral_contents', float16), ('vespene_contents', float16),
    ('cargo_space_max', uint8), ('assigned_harvesters', uint8),
    ('weapon_cooldown', uint8), ('order_length', uint8), ('order_id_0', int16),
    ('order_id_1', int16), ('is_hallucination', uint8), ('buff_id_0', uint8),
    ('buff_id_1', uint8), ('addon_unit_type', uint8), ('is_active', uint8),
    ('order_progress_0', float16), ('order_progress_1', float16),
    ('order_id_2', int16), ('order_id_3', int16), ('is_in_cargo', uint8),
    ('attack_upgrade_level', uint8), ('armor_upgrade_level', uint8),
    ('shield_upgrade_level', uint8), ('last_selected_units', int8),
    ('last_targeted_unit', int8),
]

ACTION_INFO = {
    'action_type': torch.tensor(0, dtype=torch.long),
    'delay': torch.tensor(0, dtype=torch.long),
    'queued': torch.tensor(0, dtype=torch.long),
    'selected_units': torch.zeros((MAX_SELECTED_UNITS_NUM,), dtype=torch.long),
    'target_unit': torch.tensor(0, dtype=torch.long),
    'target_location': torch.tensor(0, dtype=torch.long),
}

ACTION_LOGP = {
    'action_type': torch.tensor(0, dtype=torch.float),
    'delay': torch.tensor(0, dtype=torch.float),
    'queued': torch.tensor(0, dtype=torch.float),
    'selected_units': torch.zeros((MAX_SELECTED_UNITS_NUM,), dtype=torch.float),
    'target_unit': torch.tensor(0, dtype=torch.float),
    'target_location': torch.tensor(0, dtype=torch.float),
}

ACTION_LOGIT = {
    'action_type': torch.zeros(NUM_ACTIONS, dtype=torch.float),
    'delay': torch.zeros(MAX_DELAY + 1, dtype=torch.float),
    'queued': torch.zeros(2, dtype=torch.float),
    'selected_units': torch.zeros((MAX_SELECTED_UNITS_NUM, MAX_ENTITY_NUM + 1), dtype=torch.float),
    'target_unit': torch.zeros(MAX_ENTITY_NUM, dtype=torch.float),
    'target_location': torch.zeros(SPATIAL_SIZE[0] * SPATIAL_SIZE[1], dtype=torch.float),
}


def recursive_to_share_memory(data, batch_size):
    if isinstance(data, torch.Tensor):
        if batch_size is not None:
            data = data.repeat(batch_size, *([1] * len(data.shape)))
        return data.share_memory_()
    if isinstance(data, dict):
        return {k: recursive_to_share_memory(v, batch_size) for k, v in data.items()}
    raise TypeError(type(data))


def fake_step_data(share_memory=False, batch_size=None, train=True,
                   hidden_size=None, hidden_layer=None, entity_num=None,
                   randomize=False):
    """One synthetic observation(+labels), shaped exactly like a decoded
    replay step (reference `lib/features.py:95-133`).  With ``randomize`` the
    categorical fields are drawn uniformly from their vocabularies so that the
    batch exercises real embedding/gather paths.
    """
    gen = torch.Generator().manual_seed(0) if randomize else None

    def _rand_int(high, size, dtype):
        return torch.randint(0, max(high, 1), size=size, dtype=dtype, generator=gen)

    spatial_info, scalar_info, entity_info = {}, {}, {}
    for k, dtype in SPATIAL_INFO:
        if 'effect' in k:
            spatial_info[k] = (_rand_int(SPATIAL_SIZE[0] * SPATIAL_SIZE[1], (EFFECT_LEN,), dtype)
                               if randomize else torch.zeros(EFFECT_LEN, dtype=dtype))
        else:
            high = {'height_map': 256, 'visibility_map': 4, 'player_relative': 5}.get(k, 2)
            spatial_info[k] = (_rand_int(high, SPATIAL_SIZE, dtype)
                               if randomize else torch.zeros(size=SPATIAL_SIZE, dtype=dtype))
    for k, dtype, size in SCALAR_INFO:
        if randomize:
            high = {'home_race': 5, 'away_race': 5, 'last_queued': 2,
                    'last_delay': MAX_DELAY + 1, 'last_action_type': NUM_ACTIONS,
                    'beginning_order': 2,  # one-hot input to BO encoder is 174-dim float
                    'bo_location': SPATIAL_SIZE[0] * SPATIAL_SIZE[1]}.get(k, 2)
            if k == 'time':
                scalar_info[k] = torch.rand((), generator=gen) * 1000
            elif k in ('agent_statistics',):
                scalar_info[k] = torch.rand(size, generator=gen) * 10
            elif k == 'beginning_order':
                scalar_info[k] = _rand_int(174, size, dtype)
            else:
                scalar_info[k] = _rand_int(high, size, dtype)
        else:
            scalar_info[k] = torch.zeros(size=size, dtype=dtype)
    for k, dtype in ENTITY_INFO:
        if randomize:
            high = {'unit_type': NUM_UNIT_TYPES, 'alliance': 5, 'cargo_space_taken': 9,
                    'display_type': 5, 'x': SPATIAL_SIZE[1], 'y': SPATIAL_SIZE[0],
                    'cloak': 5, 'mineral_contents': 2, 'vespene_contents': 2,
                    'cargo_space_max': 9, 'assigned_harvesters': 24, 'weapon_cooldown': 32,
                    'order_length': 9, 'order_id_0': NUM_ACTIONS, 'order_id_1': 49,
                    'buff_id_0': 50, 'buff_id_1': 50, 'addon_unit_type': 9,
                    'order_id_2': 49, 'order_id_3': 49, 'attack_upgrade_level': 4,
                    'armor_upgrade_level': 4, 'shield_upgrade_level': 4}.get(k, 2)
            if dtype in (float16, float32):
                entity_info[k] = torch.rand((MAX_ENTITY_NUM,), generator=gen).to(dtype)
            else:
                entity_info[k] = _rand_int(high, (MAX_ENTITY_NUM,), dtype)
        else:
            entity_info[k] = torch.zeros(size=(MAX_ENTITY_NUM,), dtype=dtype)
    action_mask = {k: torch.tensor(1, dtype=torch.bool) for k in ACTION_INFO}
    if entity_num is None:
        entity_num = torch.randint(1, MAX_ENTITY_NUM, size=(), dtype=torch.long, generator=gen)
    else:
        entity_num = torch.tensor(entity_num, dtype=torch.long)
    ret = {
        'spatial_info': spatial_info,
        'scalar_info': scalar_info,
        'entity_info': entity_info,
        'entity_num': entity_num,
    }
    if train:
        action_info = copy.deepcopy(ACTION_INFO)
        if randomize:
            action_info['action_type'] = _rand_int(NUM_ACTIONS, (), torch.long)
            action_info['delay'] = _rand_int(MAX_DELAY + 1, (), torch.long)
            action_info['queued'] = _rand_int(2, (), torch.long)
            action_info['selected_units'] = _rand_int(int(entity_num), (MAX_SELECTED_UNITS_NUM,), torch.long)
            action_info['target_unit'] = _rand_int(int(entity_num), (), torch.long)
            action_info['target_location'] = _rand_int(SPATIAL_SIZE[0] * SPATIAL_SIZE[1], (), torch.long)
        su_num = torch.randint(1, MAX_SELECTED_UNITS_NUM, size=(), dtype=torch.long, generator=gen)
        ret.update({'action_info': action_info, 'action_mask': action_mask,
                    'selected_units_num': su_num})
    if share_memory:
        ret = recursive_to_share_memory(ret, batch_size)
    if hidden_size is not None:
        ret['hidden_state'] = [
            (torch.zeros(batch_size, hidden_size).share_memory_(),
             torch.zeros(batch_size, hidden_size).share_memory_())
            for _ in range(hidden_layer)
        ]
    return ret


def fake_model_output(batch_size, hidden_size, hidden_layer, teacher=False):
    """Shared-memory slab shaped like one GPU-batch-inference output
    (reference `lib/features.py:136-152`)."""
    ret = {
        'logit': copy.deepcopy(ACTION_LOGIT),
        'entity_num': torch.randint(0, MAX_ENTITY_NUM, size=(), dtype=torch.long),
        'selected_units_num': torch.randint(0, MAX_SELECTED_UNITS_NUM, size=(), dtype=torch.long),
    }
    if not teacher:
        ret.update({
            'action_info': copy.deepcopy(ACTION_INFO),
            'action_logp': copy.deepcopy(ACTION_LOGP),
            'extra_units': torch.zeros(MAX_ENTITY_NUM + 1),
        })
    ret = recursive_to_share_memory(ret, batch_size)
    ret['hidden_state'] = [
        (torch.zeros(batch_size, hidden_size).share_memory_(),
         torch.zeros(batch_size, hidden_size).share_memory_())
        for _ in range(hidden_layer)
    ]
    return ret
