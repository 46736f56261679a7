"""Generate strategy-statistics Z files from replays (reference
`distar/bin/gen_z.py`): decode the WINNING side of each replay, extract
(build order, cumulative stats, born location, end loop) via
`Features.get_z`, and aggregate into map -> race-pair -> born-location JSON.

Two sources: live replay decode (needs SC2, gated) or pre-decoded
trajectory files (--source offline), each a torch-saved
{'traj_data', 'map_name', 'race', 'opponent_race', 'born_location'} dict.
"""
import argparse
import json
import os
from collections import defaultdict

import torch

from ..lib.consts import SPATIAL_SIZE


def get_args(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument('--data', required=True, help='replay dir or decoded-step dir')
    p.add_argument('--output', default='z_out.json')
    p.add_argument('--source', default='offline', choices=['offline', 'replay'])
    p.add_argument('--min-loop', type=int, default=0)
    return p.parse_args(argv)


def aggregate(entries):
    """entries: [(map_name, mix_race, born_location, z_tuple)] -> nested dict."""
    out = defaultdict(lambda: defaultdict(lambda: defaultdict(list)))
    for map_name, mix_race, born, z in entries:
        out[map_name][mix_race][str(born)].append(z)
    return {m: {r: dict(b) for r, b in races.items()} for m, races in out.items()}


def z_from_traj(feature, traj_data, end_loop):
    bo, cum, bo_len, bo_loc = feature.get_z(traj_data)
    cum_idx = torch.nonzero(torch.as_tensor(cum) > 0).squeeze(1).tolist()
    return [bo.tolist(), cum_idx, bo_loc.tolist(), int(end_loop)]


def offline_entries(data_dir, min_loop):
    from ..lib.features import Features
    entries = []
    for name in sorted(os.listdir(data_dir)):
        path = os.path.join(data_dir, name)
        try:
            rec = torch.load(path, map_location='cpu', weights_only=False)
        except Exception as e:  # noqa: BLE001
            print(f'skip {name}: {e!r}')
            continue
        end_loop = rec.get('end_loop', 0)
        if end_loop < min_loop:
            continue

        class _F:      # minimal Features stand-in for pre-decoded data
            _beginning_order_flag = True
            _cumulative_stat_flag = True
            _filter_spine = False
            _bo_zergling_num = 8
            _zero_z_value = 1.0
            home_born_location = rec.get('born_location', 0)
            away_born_location = rec.get('away_born_location', 0)
            get_z = Features.get_z
        feature = _F()
        z = z_from_traj(feature, rec['traj_data'], end_loop)
        race = rec.get('race', 'zerg')
        opp = rec.get('opponent_race', 'zerg')
        mix = race if race == opp else race + opp
        entries.append((rec.get('map_name', 'KingsCove'), mix,
                        rec.get('born_location', 0), z))
    return entries


def replay_entries(data_dir, min_loop, cfg=None, parse_race='Z'):
    """Live replay decode (reference `bin/gen_z.py` worker_loop): decode both
    sides of each replay, keep WINNING sides of the requested race whose
    games ran at least min_loop, key by the decoder's real metadata
    (map / race pair / born location)."""
    from ..data.replay_decoder import ReplayDecoder
    from ..utils.config import Config
    decoder = ReplayDecoder(cfg or Config({}))
    entries = []
    race_initial = {'zerg': 'Z', 'terran': 'T', 'protoss': 'P', 'random': 'R'}
    for name in sorted(os.listdir(data_dir)):
        if not name.endswith('.SC2Replay'):
            continue
        for player_idx in range(2):
            traj = decoder.run(os.path.join(data_dir, name), player_idx)
            meta = decoder.last_meta
            if not traj or meta is None:
                continue
            if meta['result'] != 1:                   # winning side only
                continue
            if meta['end_loop'] < min_loop:
                continue
            if parse_race and \
                    race_initial.get(meta['home_race'], 'Z') not in parse_race:
                continue
            # Z extraction from the already-stamped steps: every step
            # carries identical beginning_order/cumulative_stat tensors
            s0 = traj[0]['scalar_info']
            cum_idx = torch.nonzero(
                torch.as_tensor(s0['cumulative_stat']) > 0).squeeze(1).tolist()
            z = [s0['beginning_order'].tolist(), cum_idx,
                 s0['bo_location'].tolist(), int(meta['end_loop'])]
            mix = meta['home_race'] if meta['home_race'] == meta['away_race'] \
                else meta['home_race'] + meta['away_race']
            entries.append((meta['map_name'], mix, meta['born_location'], z))
    decoder.close()
    return entries


def main(argv=None):
    args = get_args(argv)
    entries = offline_entries(args.data, args.min_loop) \
        if args.source == 'offline' else replay_entries(args.data, args.min_loop)
    result = aggregate(entries)
    with open(args.output, 'w') as f:
        json.dump(result, f)
    print(f'wrote {args.output}: '
          f'{sum(len(b) for races in result.values() for b in races.values())} '
          'born-location buckets')


if __name__ == '__main__':
    main()
