"""Decayed per-race telemetry aggregates for league TB logging (reference
`ctools/worker/league/{cum_stat,dist_stat,unit_num_stat}.py`):
  - DistStat: z-distance / reward telemetry per frac_id,
  - CumStat: per-z-type cumulative-stat build telemetry,
  - UnitNumStat: per-unit-count telemetry.
Plain dicts (no lambda-defaultdicts): these objects are pickled into the
league resume file.
"""
from .meters import WarmupEmaMeter


class _StatBase:
    def __init__(self, decay, warm_up_size):
        self._decay = decay
        self._warm_up_size = warm_up_size
        self._stat = {}
        self.game_count = {}

    def _meter(self, frac_id, key):
        rec = self._stat.setdefault(frac_id, {})
        if key not in rec:
            rec[key] = WarmupEmaMeter(self._decay, self._warm_up_size)
        return rec[key]

    def _bump(self, frac_id):
        self.game_count[frac_id] = self.game_count.get(frac_id, 0) + 1

    @staticmethod
    def _as_float(v):
        if isinstance(v, (str, dict, list, tuple)):
            return None
        try:
            return float(v)
        except (TypeError, ValueError):
            return None


class DistStat(_StatBase):
    not_use_keys = ['z_type', 'unit_num', 'opponent_id', 'player_id', 'race',
                    'race_id', 'winloss']

    def update(self, frac_id, stat_info):
        self._bump(frac_id)
        for k, v in stat_info.items():
            if k in self.not_use_keys:
                continue
            val = self._as_float(v)
            if val is not None and val >= 0:
                self._meter(frac_id, k).update(val)
        return True

    @property
    def stat_info_dict(self):
        return {fid: {k: m.val for k, m in rec.items()}
                for fid, rec in self._stat.items()}


class CumStat(_StatBase):
    """Per-z-type (0-3) build telemetry: out/not, out/done, in/not, in/done
    (reference cum_stat.py meaning_mapping)."""
    not_use_keys = ['z_type', 'unit_num', 'step', 'winloss', 'agent_iters',
                    'bo_reward', 'cum_reward', 'bo_len', 'dist/bo',
                    'dist/bo_location', 'dist/cum', 'opponent_id', 'player_id',
                    'race', 'race_id']
    meaning_mapping = {'out/not': 0, 'out/done': 1, 'in/not': 2, 'in/done': 3}

    def update(self, frac_id, stat_info):
        self._bump(frac_id)
        z_type = int(stat_info.get('z_type', 0))
        for k, v in stat_info.items():
            if k in self.not_use_keys:
                continue
            val = self._as_float(v)
            if val is not None and val >= 0:
                self._meter(frac_id, (k, z_type)).update(val)
        return True

    @property
    def stat_info_dict(self):
        out = {}
        for fid, rec in self._stat.items():
            agg = {}
            for (k, z), m in rec.items():
                agg.setdefault(k, {i: 0.0 for i in range(4)})[z] = m.val
            out[fid] = agg
        return out


class UnitNumStat(_StatBase):
    def update(self, frac_id, side_id, stat_info):
        self._bump(frac_id)
        for k, v in (stat_info.get('unit_num') or {}).items():
            val = self._as_float(v)
            if val is not None:
                self._meter(frac_id, k).update(val)
        return True

    @property
    def stat_info_dict(self):
        return {fid: {k: m.val for k, m in rec.items()}
                for fid, rec in self._stat.items()}
