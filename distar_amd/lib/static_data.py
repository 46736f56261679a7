"""SC2 static vocabularies and dense reorder arrays.

Raw ID lists are game data loaded from ``assets/static_data.json`` (extracted
from the reference's `distar/pysc2/lib/static_data.py`).  The *_REORDER_ARRAY
construction (sparse game id -> dense model index, with index 0 kept for
"none") follows the reference's scheme so that checkpoints and replays remain
interchangeable.
"""
import json
import os

import numpy as np
import torch

_ASSET_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), 'assets')

with open(os.path.join(_ASSET_DIR, 'static_data.json')) as _f:
    _D = json.load(_f)

UNIT_TYPES = _D['unit_types']
UPGRADES = _D['upgrades']
BUFFS = _D['buffs']
ADDON = _D['addon']
UNIT_SPECIFIC_ABILITIES = _D['unit_specific_abilities']
UNIT_GENERAL_ABILITIES = _D['unit_general_abilities']
UNIT_MIX_ABILITIES = _D['unit_mix_abilities']

NUM_UNIT_TYPES = len(UNIT_TYPES)   # 260
NUM_UPGRADES = len(UPGRADES)       # 90
NUM_BUFFS = len(BUFFS)             # 50
NUM_ADDON = len(ADDON)             # 9


def _reorder_lut(ids, offset=0):
    """Dense LUT: game id -> position (+offset); unknown ids map to 0."""
    lut = np.zeros(max(ids) + 1, dtype=np.int64)
    for pos, gid in enumerate(ids):
        lut[gid] = pos + offset
    return lut


UNIT_TYPES_REORDER_ARRAY = torch.from_numpy(_reorder_lut(UNIT_TYPES))
UPGRADES_REORDER_ARRAY = torch.from_numpy(_reorder_lut(UPGRADES))
BUFFS_REORDER_ARRAY = torch.from_numpy(_reorder_lut(BUFFS))
ADDON_REORDER_ARRAY = torch.from_numpy(_reorder_lut(ADDON))

UNIT_TYPES_REORDER = {gid: i for i, gid in enumerate(UNIT_TYPES)}
UPGRADES_REORDER = {gid: i for i, gid in enumerate(UPGRADES)}
BUFFS_REORDER = {gid: i for i, gid in enumerate(BUFFS)}
ADDON_REORDER = {gid: i for i, gid in enumerate(ADDON)}
