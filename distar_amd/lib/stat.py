"""In-game statistics tables + per-race action legality mask.

Tables come from ``assets/stat_tables.json`` (extracted from the reference's
`distar/agent/default/lib/stat.py`: `cum_dict`, `ACTION_RACE_MASK`).  The
`Stat` tracker mirrors the reference's per-unit build counts / action success
rates used for TB telemetry and play-time masking.
"""
import json
import os
from collections import defaultdict

import torch

from .actions import ACTIONS, NUM_ACTIONS

_ASSET_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), 'assets')

with open(os.path.join(_ASSET_DIR, 'stat_tables.json')) as _f:
    _T = json.load(_f)

cum_dict = _T['cum_dict']
ACTION_RACE_MASK = {race: torch.tensor(mask, dtype=torch.bool)
                    for race, mask in _T['action_race_mask'].items()}
for _race, _mask in ACTION_RACE_MASK.items():
    assert _mask.shape[0] == NUM_ACTIONS


def _unit_name_of(idx):
    """'Train_Drone_quick' -> 'Drone', 'Build_SpawningPool_pt' ->
    'SpawningPool', 'Morph_Lair_quick' -> 'Lair'."""
    name = ACTIONS[idx]['name'] if 0 <= idx < NUM_ACTIONS else ''
    parts = name.split('_')
    if len(parts) >= 2 and parts[0] in ('Train', 'Build', 'Morph', 'TrainWarp'):
        return parts[1]
    return None


class Stat:
    """Per-episode action/build statistics (reference `lib/stat.py:6-60`)."""

    def __init__(self, race='zerg'):
        self._race = race
        self.reset()

    def reset(self):
        self._action_success_count = defaultdict(int)
        self._action_count = defaultdict(int)
        self._build_count = defaultdict(int)
        self._unit_num = defaultdict(int)
        self._unit_num['max_unit_num'] = 0

    def update(self, last_action_type, action_result):
        if last_action_type is None:
            return
        idx = int(last_action_type)
        self._action_count[idx] += 1
        success = action_result in (0, 1)
        if success:
            self._action_success_count[idx] += 1
            goal = ACTIONS[idx]['goal'] if 0 <= idx < NUM_ACTIONS else 'other'
            if goal in ('build', 'unit'):
                self._build_count[idx] += 1
                name = _unit_name_of(idx)
                if name:
                    self._unit_num[name] += 1
                    self._unit_num['max_unit_num'] = max(
                        self._unit_num[name], self._unit_num['max_unit_num'])

    @property
    def unit_num(self):
        """Per-unit-name build counts + running max (reference
        `lib/stat.py:8-12,47-52` builds this from a func_id->unit-name dict;
        here the name is parsed from the action name)."""
        return dict(self._unit_num)

    def get_stat_data(self):
        data = {}
        mx = max(self._unit_num['max_unit_num'], 1)
        for k, v in self._unit_num.items():
            if k != 'max_unit_num':
                data['units/' + k] = v / mx
        for idx, count in self._action_count.items():
            name = ACTIONS[idx]['name']
            data[f'action/{name}'] = count
            data[f'action_success_rate/{name}'] = \
                self._action_success_count[idx] / max(count, 1)
        for idx, count in self._build_count.items():
            data[f'build/{ACTIONS[idx]["name"]}'] = count
        return data
