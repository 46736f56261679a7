"""Extract game-constant tables + model state-dict layout from the reference
(read-only at /root/reference) into JSON assets under distar_amd/assets/.

These are *data* contracts (SC2 ability/unit IDs, the 327-entry action table,
race legality masks, reorder arrays, checkpoint key layout), not code: the
runtime loads the JSON; no reference code is imported at runtime.
Sources (reference): distar/agent/default/lib/actions.py (ACTIONS),
lib/stat.py (cum_dict, ACTION_RACE_MASK), distar/pysc2/lib/static_data.py,
distar/agent/default/model/* (state-dict layout golden).
"""
import json
import sys
import types
import os

REF = '/root/reference'
OUT = os.path.join(os.path.dirname(os.path.abspath(__file__)), '..', 'distar_amd', 'assets')


def install_stubs():
    class _Any:
        def __getattr__(self, k):
            return _Any()

        def __call__(self, *a, **kw):
            return _Any()

    for name in ['s2clientprotocol', 's2clientprotocol.spatial_pb2', 's2clientprotocol.ui_pb2',
                 's2clientprotocol.raw_pb2', 's2clientprotocol.sc2api_pb2', 's2clientprotocol.common_pb2',
                 's2clientprotocol.error_pb2', 's2clientprotocol.debug_pb2', 's2clientprotocol.query_pb2',
                 's2clientprotocol.data_pb2', 's2clientprotocol.score_pb2',
                 'absl', 'absl.logging', 'absl.flags', 'absl.app',
                 'whichcraft', 'portpicker', 'websocket', 'sko', 'mpyq',
                 'lz4', 'lz4.frame', 'redis', 'redis.client', 'pymemcache', 'pymemcache.client',
                 'pymemcache.client.base', 'ceph', 'mc', 'tensorboardX', 'cv2',
                 'matplotlib', 'matplotlib.pyplot', 'yaml' if False else 'cv2.dummy']:
        m = types.ModuleType(name)
        m.__file__ = '<stub>'
        m.__path__ = []
        m.__getattr__ = lambda k: _Any()
        sys.modules[name] = m
    # functional minimal EasyDict (the reference's config system needs a real one)
    ed = types.ModuleType('easydict')
    ed.__file__ = '<stub>'

    class EasyDict(dict):
        def __init__(self, d=None, **kwargs):
            d = dict(d or {}, **kwargs)
            for k, v in d.items():
                setattr(self, k, v)

        def __setattr__(self, name, value):
            if isinstance(value, (list, tuple)):
                value = type(value)(self.__class__(x) if isinstance(x, dict) else x for x in value)
            elif isinstance(value, dict) and not isinstance(value, self.__class__):
                value = self.__class__(value)
            super().__setattr__(name, value)
            super().__setitem__(name, value)

        __setitem__ = __setattr__

        def __getattr__(self, name):
            try:
                return self[name]
            except KeyError:
                raise AttributeError(name)

        def update(self, e=None, **f):
            for k, v in dict(e or {}, **f).items():
                setattr(self, k, v)

        def get(self, k, default=None):
            return self[k] if k in self else default

    ed.EasyDict = EasyDict
    sys.modules['easydict'] = ed
    sys.path.insert(0, REF)


def main():
    os.makedirs(OUT, exist_ok=True)
    import torch  # noqa: F401  (import before stubs: inspect walks sys.modules)
    six_mod = types.ModuleType('torch._six')
    six_mod.__file__ = '<stub>'
    six_mod.inf = float('inf')
    sys.modules['torch._six'] = six_mod
    import numpy as np
    if not hasattr(np, 'int'):  # the 2021-era reference uses the removed np.int alias
        np.int = int
        np.float = float
        np.bool = bool
    install_stubs()

    from distar.agent.default.lib import actions as ref_actions
    with open(f'{OUT}/actions.json', 'w') as f:
        json.dump({'actions': ref_actions.ACTIONS}, f)
    print('actions:', len(ref_actions.ACTIONS))

    from distar.pysc2.lib import static_data as sd
    static = {
        'unit_types': list(sd.UNIT_TYPES),
        'upgrades': list(sd.UPGRADES),
        'buffs': list(sd.BUFFS),
        'addon': list(sd.ADDON),
        'unit_specific_abilities': list(sd.UNIT_SPECIFIC_ABILITIES),
        'unit_general_abilities': list(sd.UNIT_GENERAL_ABILITIES),
        'unit_mix_abilities': list(sd.UNIT_MIX_ABILITIES),
    }
    with open(f'{OUT}/static_data.json', 'w') as f:
        json.dump(static, f)
    print('static_data:', {k: len(v) for k, v in static.items()})

    from distar.agent.default.lib import stat as ref_stat
    out = {
        'cum_dict': ref_stat.cum_dict,
        'action_race_mask': {race: mask.long().tolist() for race, mask in ref_stat.ACTION_RACE_MASK.items()},
    }
    with open(f'{OUT}/stat_tables.json', 'w') as f:
        json.dump(out, f)
    print('stat tables ok; races:', list(out['action_race_mask'].keys()))

    # Golden checkpoint layout: key -> shape for both model variants.
    from distar.agent.default.model.model import Model
    from easydict import EasyDict  # the stub installed above
    golden = {}
    m = Model(cfg={'common': {'type': 'train'}}, use_value_network=False)
    golden['policy'] = {k: list(v.shape) for k, v in m.state_dict().items()}
    cfg = EasyDict({'learner': {'use_value_feature': True}, 'common': {'type': 'train'}})
    m2 = Model(cfg=cfg, use_value_network=True)
    golden['value'] = {k: list(v.shape) for k, v in m2.state_dict().items()}
    with open(f'{OUT}/ckpt_layout_golden.json', 'w') as f:
        json.dump(golden, f, indent=0)
    print('golden keys:', len(golden['policy']), len(golden['value']))

    # raw action-function table: func_id -> ability/general/type (data for
    # reverse_raw_action and transform_action)
    from distar.pysc2.lib import actions as pysc2_actions
    raw_funcs = []
    for f in pysc2_actions.RAW_FUNCTIONS:
        raw_funcs.append({
            'id': int(f.id), 'name': f.name,
            'ability_id': int(f.ability_id) if f.ability_id else 0,
            'general_id': int(f.general_id) if f.general_id else 0,
            'function_type': f.function_type.__name__,
        })
    with open(f'{OUT}/raw_functions.json', 'w') as fo:
        json.dump({'raw_functions': raw_funcs}, fo)
    print('raw functions:', len(raw_funcs))


if __name__ == '__main__':
    main()

# Note: the strategy-statistics Z files (distar_amd/assets/z_files/*.json) are
# replay-derived *data* copied from the reference's lib/ directory — the same
# artifacts our bin/gen_z.py regenerates from replay packs.
