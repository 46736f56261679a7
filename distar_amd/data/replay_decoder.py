"""Replay decoder: .SC2Replay -> supervised training step lists.

Functional parity with the reference's
`agent/default/replay_decoder.py:216-435`: per replay+player,
  pass 1 at 1x1 resolution steps 50 loops at a time harvesting raw actions
  with `FilterActions` de-duplication of spammed train/morph/research
  commands,
  pass 2 re-opens at map resolution, steps the controller by inter-action
  delays, runs `Features.transform_obs` + `reverse_raw_action` per action,
  appends Z statistics to every step, restarts SC2 every 10 replays and on
  parse errors, with SC2-version routing from replay metadata.

Needs a StarCraft II install + s2clientprotocol (gated like envs/env.py);
`FilterActions` and the windowing logic are pure and unit-tested on CPU.
"""
import json
import os

import torch

from ..envs.protocol import (SC2_PROTO_AVAILABLE, RemoteController,
                             launch_game_process, version_info)
from ..lib.consts import MAX_DELAY
from ..lib.features import Features
from ..lib.mpq import MPQArchive

RESTART_REPLAY_INTERVAL = 10
PASS1_STEP = 50


class FilterActions:
    """De-duplicate spammed commands (reference `replay_decoder.py:70-213`):
    repeated identical train/research/morph commands within a short window
    collapse to one action."""

    TRAIN_WINDOW = 3        # game loops x PASS1 step granularity

    def __init__(self, cfg=None):
        from ..lib.actions import ACTIONS, FUNC_ID_TO_ACTION_TYPE_DICT
        self._filter_gabs = set()
        for a in ACTIONS:
            if a['goal'] in ('unit', 'research') or 'Morph' in a['name']:
                self._filter_gabs.add(a['general_ability_id'])

    def run(self, actions_with_loops):
        """[(game_loop, ability_id, unit_tags, action_obj)] -> filtered list.
        Consecutive identical (ability, tags) pairs within TRAIN_WINDOW loops
        keep only the last occurrence."""
        out = []
        for i, (loop, ab, tags, act) in enumerate(actions_with_loops):
            if ab in self._filter_gabs and out:
                ploop, pab, ptags, _ = out[-1]
                if pab == ab and ptags == tags and loop - ploop <= self.TRAIN_WINDOW:
                    out[-1] = (loop, ab, tags, act)
                    continue
            out.append((loop, ab, tags, act))
        return out


class ReplayDecoder:
    def __init__(self, cfg):
        if not SC2_PROTO_AVAILABLE:      # pragma: no cover - protobuf absent
            raise ImportError('ReplayDecoder needs google.protobuf')
        self._whole_cfg = cfg
        self._filter = FilterActions(cfg)
        self._proc = None
        self._controller = None
        self._decode_count = 0
        self._cur_version = None
        # metadata of the last successful decode (gen_z consumes this):
        # {'map_name','home_race','away_race','born_location','result',
        #  'end_loop'}
        self.last_meta = None

    def _version_of(self, replay_path):
        """SC2 version sniff from the replay's MPQ `replay.gamemetadata.json`
        (reference :361-380).  Returns a '4.10.0'-style string the version
        table (assets/sc2_versions.json) can route to a Base<build> binary,
        or None when the metadata is unreadable/unknown."""
        try:
            meta = json.loads(
                MPQArchive(replay_path).read_file('replay.gamemetadata.json'))
            version = '.'.join(meta['GameVersion'].split('.')[:3])
            base_build = int(meta['BaseBuild'][4:])
            if version_info(version) is None:
                # unknown point release: route by the base build directly
                return base_build
            return version
        except Exception:  # noqa: BLE001
            return None

    def _ensure_sc2(self, version):
        if self._controller is not None and version == self._cur_version and \
                self._decode_count % RESTART_REPLAY_INTERVAL != 0:
            return
        self.close()
        # the sniffed version selects the binary + -dataVersion
        self._proc, port = launch_game_process(self._whole_cfg,
                                               version=version)
        self._controller = RemoteController('127.0.0.1', port)
        self._cur_version = version

    def run(self, replay_path, player_idx):
        try:
            return self._parse_replay(replay_path, player_idx)
        except Exception as e:  # noqa: BLE001 - decoder restarts on any error
            print(f'[ReplayDecoder] {replay_path} failed: {e!r}')
            self.close()
            return None
        finally:
            self._decode_count += 1

    def _parse_replay(self, replay_path, player_idx):
        version = self._version_of(replay_path)
        self._ensure_sc2(version)
        ctrl = self._controller
        # pass 1: harvest actions at 1x1 minimap resolution
        ctrl.start_replay(replay_path, player_idx + 1,
                          minimap_resolution=(1, 1))
        game_info = ctrl.game_info()
        map_size = (game_info.start_raw.map_size.x,
                    game_info.start_raw.map_size.y)
        raw_actions = []
        outcome = None
        end_loop = 0
        while True:
            obs = ctrl.observe()
            end_loop = obs['game_loop']
            for act in getattr(obs['raw_obs'], 'actions', []):
                if act.HasField('action_raw'):
                    uc = act.action_raw.unit_command \
                        if act.action_raw.HasField('unit_command') else None
                    raw_actions.append((obs['game_loop'],
                                        uc.ability_id if uc else 0,
                                        tuple(uc.unit_tags) if uc else (), act))
            results = getattr(obs['raw_obs'], 'player_result', [])
            if results:
                # the OBSERVED player's result (player_result lists all)
                raw = next((r.result for r in results
                            if r.player_id == player_idx + 1),
                           results[0].result)
                outcome = {1: 1, 2: -1, 3: 0}.get(raw, 0)
                break
            ctrl.step(PASS1_STEP)
        raw_actions = self._filter.run(raw_actions)
        # pass 2: re-open at map resolution, step by inter-action delays,
        # transform each action (reference :279-335)
        ctrl.start_replay(replay_path, player_idx + 1,
                          minimap_resolution=map_size)
        feature = Features(ctrl.game_info(), ctrl.observe()['raw_obs'],
                           self._whole_cfg)
        traj_data = []
        prev_loop = 0
        last_sel, last_tar = None, None
        for loop, _, _, act in raw_actions:
            delay = min(loop - prev_loop, MAX_DELAY)
            if delay > 0:
                ctrl.step(delay)
            prev_loop = loop
            obs = ctrl.observe()
            step = feature.transform_obs(obs['raw_obs'], padding_spatial=False)
            (action_info, action_mask, su_num, last_sel, last_tar, invalid) = \
                feature.reverse_raw_action(act, step['game_info']['tags'])
            if invalid:
                continue
            action_info['delay'] = torch.tensor(delay, dtype=torch.long)
            step.pop('game_info')
            step.update({'action_info': action_info, 'action_mask': action_mask,
                         'selected_units_num': su_num})
            traj_data.append(step)
        # append Z statistics to every step (reference :337-348)
        bo, cum, bo_len, bo_loc = feature.get_z(traj_data)
        for step in traj_data:
            step['scalar_info']['beginning_order'] = bo
            step['scalar_info']['bo_location'] = bo_loc
            step['scalar_info']['cumulative_stat'] = cum
        # decode metadata for gen_z (map, races, born location, outcome of
        # the observed player — reference gen_z keeps winning sides only)
        race_names = {1: 'terran', 2: 'zerg', 3: 'protoss', 4: 'random'}
        races = {info.player_id: race_names.get(int(info.race_requested), 'zerg')
                 for info in game_info.player_info}
        home_id = player_idx + 1
        away_id = next((pid for pid in races if pid != home_id), home_id)
        self.last_meta = {
            'map_name': game_info.map_name,
            'home_race': races.get(home_id, 'zerg'),
            'away_race': races.get(away_id, 'zerg'),
            'born_location': getattr(feature, 'home_born_location', 0),
            'result': outcome,
            'end_loop': end_loop,
        }
        return traj_data

    def close(self):
        if self._controller is not None:
            try:
                self._controller.quit()
            except Exception:  # noqa: BLE001
                pass
        if self._proc is not None:
            try:
                self._proc.kill()
            except Exception:  # noqa: BLE001
                pass
        self._controller = None
        self._proc = None
