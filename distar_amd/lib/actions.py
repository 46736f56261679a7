"""The 327-entry action table and every set derived from it.

The table itself is *game data* (SC2 func/ability IDs and per-head
applicability flags), loaded from ``distar_amd/assets/actions.json`` which was
extracted from the reference (`distar/agent/default/lib/actions.py:5-397`).
All derived structures (`SELECTED_UNITS_MASK`, `QUEUE_ACTIONS`,
`BEGINNING_ORDER_ACTIONS`, `CUMULATIVE_STAT_ACTIONS`, `UNIT_TO_CUM`, ...)
follow the reference's derivation rules (`lib/actions.py:398-426`).
"""
import json
import os
from collections import defaultdict

import torch

from .static_data import (UNIT_SPECIFIC_ABILITIES, UNIT_GENERAL_ABILITIES,
                          UNIT_MIX_ABILITIES)

_ASSET_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), 'assets')

with open(os.path.join(_ASSET_DIR, 'actions.json')) as _f:
    ACTIONS = json.load(_f)['actions']

NUM_ACTIONS = len(ACTIONS)  # 327
NUM_UNIT_MIX_ABILITIES = len(UNIT_MIX_ABILITIES)  # 269

ACTIONS_BY_NAME = {a['name']: idx for idx, a in enumerate(ACTIONS)}
FUNC_ID_TO_ACTION_TYPE_DICT = {a['func_id']: idx for idx, a in enumerate(ACTIONS)}

# ability id -> general ability id (identity when no general id)
ABILITY_TO_GABILITY = {}
for _i, _sab in enumerate(UNIT_SPECIFIC_ABILITIES):
    _gab = UNIT_GENERAL_ABILITIES[_i]
    ABILITY_TO_GABILITY[_sab] = _sab if _gab == 0 else _gab

# ability id -> index into the mixed-ability vocabulary (reference
# `lib/actions.py:339-347`); 0 is reserved for no-op.
UNIT_ABILITY_REORDER = torch.full((max(UNIT_MIX_ABILITIES) + 1,), -1, dtype=torch.long)
for _i, _sab in enumerate(UNIT_SPECIFIC_ABILITIES):
    _gab = UNIT_GENERAL_ABILITIES[_i]
    _target = _sab if _gab == 0 else _gab
    UNIT_ABILITY_REORDER[_sab] = UNIT_MIX_ABILITIES.index(_target)
UNIT_ABILITY_REORDER[0] = 0

# Train_/Research actions form the "queue action" vocabulary (49 + no-op).
GABILITY_TO_QUEUE_ACTION = {}
QUEUE_ACTIONS = []
_count = 1  # 0 = no-op
for _idx, _a in enumerate(ACTIONS):
    if 'Train_' in _a['name'] or 'Research' in _a['name']:
        GABILITY_TO_QUEUE_ACTION[_a['general_ability_id']] = _count
        QUEUE_ACTIONS.append(_idx)
        _count += 1
    else:
        GABILITY_TO_QUEUE_ACTION[_a['general_ability_id']] = 0

ABILITY_TO_QUEUE_ACTION = torch.full((max(ABILITY_TO_GABILITY.keys()) + 1,), -1, dtype=torch.long)
ABILITY_TO_QUEUE_ACTION[0] = 0
for _aid, _gid in ABILITY_TO_GABILITY.items():
    ABILITY_TO_QUEUE_ACTION[_aid] = GABILITY_TO_QUEUE_ACTION.get(_gid, 0)

# Strategy-statistics vocabularies (reference `lib/actions.py:375-400`).
EXCLUDE_ACTIONS = [
    'Build_Pylon_pt', 'Train_Overlord_quick', 'Build_SupplyDepot_pt',   # supply
    'Train_Drone_quick', 'Train_SCV_quick', 'Train_Probe_quick',        # workers
    'Build_CreepTumor_pt', '',
]
CUM_EXCLUDE_ACTIONS = [
    'Build_SpineCrawler_pt', 'Build_SporeCrawler_pt', 'Build_PhotonCannon_pt',
    'Build_ShieldBattery_pt', 'Build_Bunker_pt', 'Morph_Overseer_quick',
    'Build_MissileTurret_pt',
]

BEGINNING_ORDER_ACTIONS = [0]
CUMULATIVE_STAT_ACTIONS = [0]
for _idx, _a in enumerate(ACTIONS):
    if _a['goal'] in ('unit', 'build', 'research') and _a['name'] not in EXCLUDE_ACTIONS:
        BEGINNING_ORDER_ACTIONS.append(_idx)
        if _a['name'] not in CUM_EXCLUDE_ACTIONS:
            CUMULATIVE_STAT_ACTIONS.append(_idx)

NUM_QUEUE_ACTIONS = len(QUEUE_ACTIONS)                        # 49
NUM_BEGINNING_ORDER_ACTIONS = len(BEGINNING_ORDER_ACTIONS)    # 174
NUM_CUMULATIVE_STAT_ACTIONS = len(CUMULATIVE_STAT_ACTIONS)    # 167

SELECTED_UNITS_MASK = torch.tensor([bool(a['selected_units']) for a in ACTIONS], dtype=torch.bool)

UNIT_BUILD_ACTIONS = [a['func_id'] for a in ACTIONS if a['goal'] == 'build']
UNIT_TRAIN_ACTIONS = [a['func_id'] for a in ACTIONS if a['goal'] == 'unit']

GENERAL_ABILITY_IDS = [a['general_ability_id'] for a in ACTIONS]
UNIT_ABILITY_TO_ACTION = {}
for _idx, _ab in enumerate(UNIT_MIX_ABILITIES):
    if _ab in GENERAL_ABILITY_IDS:
        UNIT_ABILITY_TO_ACTION[_idx] = GENERAL_ABILITY_IDS.index(_ab)

# game unit/upgrade id -> cumulative-stat slot
UNIT_TO_CUM = defaultdict(lambda: -1)
UPGRADE_TO_CUM = defaultdict(lambda: -1)
for _idx, _a in enumerate(ACTIONS):
    if 'game_id' not in _a or _idx not in CUMULATIVE_STAT_ACTIONS:
        continue
    if _a['goal'] in ('unit', 'build'):
        UNIT_TO_CUM[_a['game_id']] = CUMULATIVE_STAT_ACTIONS.index(_idx)
    elif _a['goal'] == 'research':
        UPGRADE_TO_CUM[_a['game_id']] = CUMULATIVE_STAT_ACTIONS.index(_idx)
