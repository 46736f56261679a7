"""Rollout agent: obs transform + model inference + trajectory assembly.

Functional parity with the reference's `distar/agent/default/agent.py:92-775`
and the `docs/agent.md` contract:
  - ``reset(map_name, race, opponent_race, obs)``: sample a Z (strategy
    statistics) for the born location, set pseudo-reward targets,
  - ``step(observation) -> [action_dict]`` with
    {func_id, skip_steps, queued, unit_tags, target_unit_tag, location},
  - ``collect_data(next_obs, reward, done, idx)``: teacher-KL logits,
    build-order (Levenshtein) / cumulative-stat (Hamming) / battle
    pseudo-rewards, per-head masks, traj_len-chunked step lists,
  - stat/telemetry accessors (get_stat_data / get_behavior_z / ...).

Observation modes: 'mock' consumes the tensor observations MockSC2Env emits
(synthetic rollouts, CPU tests); 'sc2' transforms raw protobufs through
`lib.features.Features` (gated on s2clientprotocol).
"""
import copy
import json
import os
import random
from collections import defaultdict, deque
from functools import partial

import torch

from ..lib.actions import (ACTIONS, BEGINNING_ORDER_ACTIONS,
                           CUMULATIVE_STAT_ACTIONS,
                           NUM_CUMULATIVE_STAT_ACTIONS, QUEUE_ACTIONS,
                           UNIT_ABILITY_TO_ACTION, UNIT_TO_CUM,
                           UPGRADE_TO_CUM)
from ..lib.consts import (BEGINNING_ORDER_LENGTH, MAX_DELAY, SPATIAL_SIZE)
from ..lib.stat import Stat, cum_dict
from ..models.alphastar.model import Model
from ..utils.data import default_collate_with_dim, to_device
from ..utils.metric import hamming_distance, l2_distance, levenshtein_distance
from ..utils.timing import sw

Z_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                     'assets', 'z_files')
RACE_DICT = {1: 'terran', 2: 'zerg', 3: 'protoss', 4: 'random'}
DEFAULT_AGENT_CFG = {
    'z_path': '3map.json', 'fake_reward_prob': 1.0, 'clip_bo': False,
    'cum_type': 'action', 'zero_z_exceed_loop': True, 'zero_z_value': 1.0,
    'extra_units': False, 'battle_norm': 30,
}


class Agent:
    HAS_MODEL = True
    HAS_TEACHER = True

    def __init__(self, cfg=None, env_id=0):
        self._whole_cfg = cfg or {}
        agent_cfg = dict(DEFAULT_AGENT_CFG)
        agent_cfg.update((cfg or {}).get('agent', {}))
        self._cfg = agent_cfg
        self._env_id = env_id
        self._job_type = (cfg or {}).get('actor', {}).get('job_type', 'train')
        self._use_value_feature = (cfg or {}).get('learner', {}).get('use_value_feature', False)
        self._traj_len = (cfg or {}).get('actor', {}).get('traj_len', 16)
        self._use_cuda = (cfg or {}).get('actor', {}).get('use_cuda', False)
        self.model = Model(self._whole_cfg or {'common': {'type': 'train'}})
        self.model.eval()
        self.teacher_model = None
        self.successive_model = None    # DAPO (reference agent.py:506-515)
        self.z_idx = None               # play-ckpt curated Z subset (agent.py:119,202-204)
        self.player_id = 'MP0'
        self.race = 'zerg'
        self._num_layers = self.model.cfg.encoder.core_lstm.num_layers
        self._hidden_size = self.model.cfg.encoder.core_lstm.hidden_size
        self._stat_api = Stat('zerg')
        self._model_last_iter = 0
        self._data_buffer = deque(maxlen=self._traj_len)
        self._push_count = 0
        self._iter_count = 0

    # ------------------------------------------------------------------ reset
    def reset(self, map_name='KingsCove', race='zerg', opponent_race='zerg',
              obs=None):
        self._map_name = map_name
        self.race = race
        self._stat_api = Stat(race)
        self._iter_count = 0
        self._data_buffer = deque(maxlen=self._traj_len)
        self._push_count = 0
        z = torch.zeros
        self._hidden_state = [(z(self._hidden_size), z(self._hidden_size))
                              for _ in range(self._num_layers)]
        self._teacher_hidden_state = [(z(self._hidden_size), z(self._hidden_size))
                                      for _ in range(self._num_layers)]
        self._successive_hidden_state = [(z(self._hidden_size), z(self._hidden_size))
                                         for _ in range(self._num_layers)]
        self._hidden_state_backup = self._hidden_state
        self._last_action_type = torch.tensor(0, dtype=torch.long)
        self._last_delay = torch.tensor(0, dtype=torch.long)
        self._last_queued = torch.tensor(0, dtype=torch.long)
        self._last_location = torch.tensor(0, dtype=torch.long)
        self._last_selected_unit_tags = None
        self._last_target_unit_tag = None
        self._enemy_unit_type_bool = torch.zeros(260, dtype=torch.uint8)
        self._exceed_flag = True
        self._bo_zergling_count = 0
        self._game_step = 0
        self._behaviour_building_order = []
        self._behaviour_bo_location = []
        self._behaviour_cumulative_stat = [0] * NUM_CUMULATIVE_STAT_ACTIONS
        self._total_bo_reward = torch.zeros((), dtype=torch.float)
        self._total_cum_reward = torch.zeros((), dtype=torch.float)
        self._game_info = {'battle_score': 0, 'opponent_battle_score': 0}
        # own first-base raw position for the observation-mode cum skip
        # (reference agent.py:176-183)
        self._born_location_xy = None
        raw = (obs or {}).get('raw_obs') if isinstance(obs, dict) else None
        if raw is not None:
            try:
                locs = [[u.pos.x, u.pos.y]
                        for u in raw.observation.raw_data.units
                        if u.unit_type in (59, 18, 86)]
                if locs:
                    self._born_location_xy = locs[0]
            except AttributeError:
                pass
        self._load_z(map_name, race, opponent_race, obs)
        self._old_bo_reward = -levenshtein_distance(
            torch.as_tensor(self._behaviour_building_order, dtype=torch.long),
            self._target_building_order) / self._bo_norm
        self._old_cum_reward = -hamming_distance(
            torch.as_tensor(self._behaviour_cumulative_stat, dtype=torch.float),
            self._target_cumulative_stat) / self._cum_norm

    def _load_z(self, map_name, race, opponent_race, obs):
        """Sample a strategy-statistics Z for this map/race/born-location
        (reference agent.py:176-243)."""
        z_path = os.path.join(Z_DIR, os.path.basename(str(self._cfg['z_path'])))
        z_data = {}
        if os.path.isfile(z_path):
            with open(z_path) as f:
                z_data = json.load(f)
        mix_race = race if race == opponent_race else race + opponent_race
        entry = None
        map_entry = z_data.get(map_name) or next(iter(z_data.values()), {})
        race_entry = map_entry.get(mix_race) or map_entry.get(race) or \
            next(iter(map_entry.values()), {})
        z_type = None
        if race_entry:
            born = self._born_location_key(obs, race_entry)
            zs = race_entry.get(born) or next(iter(race_entry.values()))
            if self.z_idx is not None:
                # curated subset shipped inside a play checkpoint (reference
                # agent.py:202-204): pick (index, z_type) pairs
                curated = self.z_idx.get(map_name, {}).get(mix_race, {}) \
                    .get(str(born)) if isinstance(self.z_idx, dict) else None
                if curated:
                    idx, z_type = random.choice(curated)
                    entry = zs[idx]
                else:
                    entry = random.choice(zs)
            else:
                entry = random.choice(zs)
        if entry is not None:
            if len(entry) == 5:
                bo, cum, bo_loc, self._target_z_loop, z_type = entry
            else:
                bo, cum, bo_loc, self._target_z_loop = entry
        else:
            bo, cum, bo_loc, self._target_z_loop = [0] * 20, [0], [0] * 20, 1 << 30
        self.use_bo_reward = True
        self.use_cum_reward = True
        if z_type in (2, 3):
            self.use_cum_reward = False
        if z_type in (1, 3):
            self.use_bo_reward = False
        if random.random() > self._cfg['fake_reward_prob']:
            self.use_cum_reward = False
        if random.random() > self._cfg['fake_reward_prob']:
            self.use_bo_reward = False
        self._bo_norm = max(len(bo), 1)
        self._cum_norm = max(len(cum), 1)
        self._target_building_order = torch.as_tensor(bo, dtype=torch.long)
        self._target_bo_location = torch.as_tensor(bo_loc, dtype=torch.long)
        self._target_cumulative_stat = torch.zeros(NUM_CUMULATIVE_STAT_ACTIONS)
        self._target_cumulative_stat.scatter_(
            0, torch.as_tensor(cum, dtype=torch.long), 1.)

    @staticmethod
    def _born_location_key(obs, race_entry):
        raw = (obs or {}).get('raw_obs') if isinstance(obs, dict) else None
        if raw is not None:
            for u in raw.observation.raw_data.units:
                if u.unit_type in (59, 18, 86):
                    x, y = int(u.pos.x), int(u.pos.y)
                    return str(x + y * SPATIAL_SIZE[1])
        return random.choice(list(race_entry.keys()))

    # ------------------------------------------------------------------ step
    def _transform_obs(self, obs):
        if obs.get('raw_obs') is not None:
            from ..lib.features import Features
            if not hasattr(self, '_feature'):
                self._feature = Features(obs['game_info_proto'], obs['raw_obs'],
                                         self._whole_cfg)
            agent_obs = self._feature.transform_obs(
                obs['raw_obs'], padding_spatial=True,
                opponent_obs=obs.get('opponent_obs') if self._use_value_feature else None)
        else:
            agent_obs = {k: copy.deepcopy(obs[k]) for k in
                         ('spatial_info', 'scalar_info', 'entity_info', 'entity_num')}
            en = int(agent_obs['entity_num'])
            agent_obs['game_info'] = {'game_loop': obs.get('game_loop', 0),
                                      'tags': list(range(en)),
                                      'battle_score': 0, 'opponent_battle_score': 0}
        return agent_obs

    def _pre_process(self, obs):
        agent_obs = self._transform_obs(obs)
        game_info = agent_obs.pop('game_info')
        self._game_info.update(game_info)
        self._game_step = game_info['game_loop']
        if self._cfg['zero_z_exceed_loop'] and self._game_step > self._target_z_loop:
            self._exceed_flag = False
            self._target_z_loop = 99999999
        en = int(agent_obs['entity_num'])
        lsu = torch.zeros(en, dtype=torch.int8)
        ltu = torch.zeros(en, dtype=torch.int8)
        tags = game_info['tags']
        if self._last_selected_unit_tags:
            for t in self._last_selected_unit_tags:
                if t in tags:
                    lsu[tags.index(t)] = 1
        if self._last_target_unit_tag is not None and self._last_target_unit_tag in tags:
            ltu[tags.index(self._last_target_unit_tag)] = 1
        pad = agent_obs['entity_info']['x'].shape[0] - en
        agent_obs['entity_info']['last_selected_units'] = \
            torch.nn.functional.pad(lsu, (0, pad))
        agent_obs['entity_info']['last_targeted_unit'] = \
            torch.nn.functional.pad(ltu, (0, pad))
        si = agent_obs['scalar_info']
        si['last_delay'] = self._last_delay.clamp(max=MAX_DELAY)
        si['last_action_type'] = self._last_action_type
        si['last_queued'] = self._last_queued
        si['enemy_unit_type_bool'] = (self._enemy_unit_type_bool |
                                      si['enemy_unit_type_bool'].to(torch.uint8))
        self._enemy_unit_type_bool = si['enemy_unit_type_bool']
        gate = bool(self.use_bo_reward and self._exceed_flag)
        si['beginning_order'] = self._target_building_order * gate
        si['bo_location'] = self._target_bo_location * gate
        if self.use_cum_reward and self._exceed_flag:
            si['cumulative_stat'] = self._target_cumulative_stat
        else:
            si['cumulative_stat'] = self._target_cumulative_stat * 0 + \
                self._cfg['zero_z_value']
        agent_obs['hidden_state'] = self._hidden_state
        self._observation = agent_obs
        model_input = default_collate_with_dim([
            {k: v for k, v in agent_obs.items() if k != 'hidden_state'}])
        model_input['hidden_state'] = [(h.unsqueeze(0), c.unsqueeze(0))
                                       for h, c in self._hidden_state]
        if self._use_cuda:
            model_input = to_device(model_input, 'cuda')
        return model_input

    def decollate_output(self, output, k=None):
        if isinstance(output, torch.Tensor):
            return output.squeeze(0).cpu()
        if k == 'hidden_state':
            return [(output[l][0].squeeze(0).cpu(), output[l][1].squeeze(0).cpu())
                    for l in range(len(output))]
        if isinstance(output, dict):
            return {key: self.decollate_output(v, key) for key, v in output.items()}
        return output

    def attach_batch_inference(self, server, env_id):
        """Switch to shared-slab batched inference (reference
        agent.py:128-141): obs go into the server's input slab; outputs are
        read back per slot after the server tick."""
        self._batch_server = server
        self._env_id = env_id

    def _batched_infer(self, agent_obs):
        import time as _time
        from .batch_inference import copy_input_data
        server = self._batch_server
        copy_input_data(server.shared_input, agent_obs, data_idx=self._env_id)
        server.signals[self._env_id] += 1
        while int(server.signals[self._env_id]) != 0:
            _time.sleep(0.001)
        out = server.shared_output
        idx = self._env_id
        result = {
            'action_info': {k: v[idx].clone() for k, v in out['action_info'].items()},
            'action_logp': {k: v[idx].clone() for k, v in out['action_logp'].items()},
            'logit': {k: v[idx].clone() for k, v in out['logit'].items()},
            'selected_units_num': out['selected_units_num'][idx].clone(),
            'entity_num': out['entity_num'][idx].clone(),
            'extra_units': out['extra_units'][idx].clone(),
            'hidden_state': [(out['hidden_state'][l][0][idx].clone(),
                              out['hidden_state'][l][1][idx].clone())
                             for l in range(len(out['hidden_state']))],
        }
        en, su = int(result['entity_num']), int(result['selected_units_num'])
        result['logit']['selected_units'] = \
            result['logit']['selected_units'][:max(su, 1), :en + 1]
        result['logit']['target_unit'] = result['logit']['target_unit'][:en]
        result['action_info']['selected_units'] = \
            result['action_info']['selected_units'][:max(su, 1)]
        result['action_logp']['selected_units'] = \
            result['action_logp']['selected_units'][:max(su, 1)]
        return result

    def _batched_teacher_logit(self, teacher_obs):
        """Teacher KL through the server's teacher slab (reference
        agent.py:715-739): write obs + recurrent state at our slot, signal,
        poll, read back the trimmed logits."""
        import time as _time
        from .batch_inference import copy_input_data
        server = self._batch_server
        idx = self._env_id
        payload = dict(teacher_obs)
        payload['hidden_state'] = self._teacher_hidden_state
        copy_input_data(server.teacher_input, payload, data_idx=idx)
        server.teacher_signals[idx] += 1
        while int(server.teacher_signals[idx]) != 0:
            _time.sleep(0.001)
        out = server.teacher_output
        result = {
            'logit': {k: v[idx].clone() for k, v in out['logit'].items()},
            'entity_num': out['entity_num'][idx].clone(),
            'selected_units_num': out['selected_units_num'][idx].clone(),
            'hidden_state': [(out['hidden_state'][l][0][idx].clone(),
                              out['hidden_state'][l][1][idx].clone())
                             for l in range(len(out['hidden_state']))],
        }
        en = int(teacher_obs['entity_num'])
        su = int(teacher_obs['selected_units_num'])
        result['logit']['selected_units'] = \
            result['logit']['selected_units'][:max(su, 1), :en + 1]
        result['logit']['target_unit'] = result['logit']['target_unit'][:en]
        return result

    @sw.decorate('agent_step')
    def step(self, observation):
        if 'eval' in self._job_type and self._iter_count > 0 and \
                not self._whole_cfg.get('env', {}).get('realtime', False):
            self._update_fake_reward(int(self._last_action_type),
                                     self._last_location, observation)
        model_input = self._pre_process(observation)
        self._stat_api.update(int(self._last_action_type),
                              (observation.get('action_result') or [1])[0])
        if getattr(self, '_batch_server', None) is not None:
            obs_for_slab = dict(self._observation)
            output = self._batched_infer(obs_for_slab)
            action = self._post_process(output)
            self._iter_count += 1
            return action
        with torch.no_grad():
            model_output = self.model.compute_logp_action(**model_input)
        action = self._post_process(self.decollate_output(model_output))
        self._iter_count += 1
        return action

    def _post_process(self, output):
        self._hidden_state = output['hidden_state']
        self._last_queued = output['action_info']['queued']
        self._last_action_type = output['action_info']['action_type']
        self._last_delay = output['action_info']['delay']
        self._last_location = output['action_info']['target_location']
        self._output = output
        at = int(output['action_info']['action_type'])
        tags = self._game_info['tags']
        action_info = {
            'func_id': ACTIONS[at]['func_id'],
            'skip_steps': int(output['action_info']['delay']),
            'queued': int(output['action_info']['queued']),
            'unit_tags': [],
        }
        su_num = int(output['selected_units_num'])
        for i in range(max(su_num - 1, 0)):
            idx = int(output['action_info']['selected_units'][i])
            if idx < len(tags):
                action_info['unit_tags'].append(tags[idx])
        if self._cfg['extra_units'] and 'extra_units' in output:
            for idx in torch.nonzero(output['extra_units']).squeeze(1).tolist():
                if idx < len(tags):
                    action_info['unit_tags'].append(tags[idx])
        self._last_selected_unit_tags = \
            action_info['unit_tags'] if ACTIONS[at]['selected_units'] else None
        tu = int(output['action_info']['target_unit'])
        action_info['target_unit_tag'] = tags[tu] if tu < len(tags) else 0
        self._last_target_unit_tag = \
            action_info['target_unit_tag'] if ACTIONS[at]['target_unit'] else None
        loc = int(output['action_info']['target_location'])
        x, y = loc % SPATIAL_SIZE[1], loc // SPATIAL_SIZE[1]
        # model space -> game world coordinates: transform_obs maps
        # y_world -> map_size.y - y, so invert on the way out (reference
        # agent.py:389-391); the mock env has no real coordinate frame
        if hasattr(self, '_feature'):
            y = max(self._feature.map_size.y - y, 0)
        action_info['location'] = (x, y)
        if 'test' in self._job_type:
            self._print_action(output['action_info'], (x, y),
                               output.get('action_logp', {}))
        return [action_info]

    def _print_action(self, action_info, location, logp):
        """Human-readable action trace for eval/test runs (reference
        agent.py:398-417)."""
        at = int(action_info['action_type'])
        name = ACTIONS[at]['name']
        def p(head):
            v = logp.get(head)
            return f'{float(torch.exp(v)):.2f}' if v is not None else '?'
        su = ''
        if ACTIONS[at]['selected_units'] and 'selected_units' in action_info:
            sel = action_info['selected_units']
            lp = logp.get('selected_units')
            for i, u in enumerate(sel[:-1].tolist()):
                prob = f'({float(torch.exp(lp[i])):.2f})' if lp is not None else ''
                su += f' {int(self._observation["entity_info"]["unit_type"][u])}{prob}'
            if lp is not None and len(lp):
                su += f' end({float(torch.exp(lp[-1])):.2f})'
        tu = int(action_info['target_unit']) if ACTIONS[at]['target_unit'] else None
        print(f'{self.player_id}, game_step:{self._game_step}, '
              f'at:{name}({p("action_type")}), '
              f'delay:{int(action_info["delay"])}({p("delay")}), su:{su}, '
              f'tu:{tu}({p("target_unit")}), '
              f'lo:{location}({p("target_location")})')

    # ----------------------------------------------------------- collect_data
    def collect_data(self, next_obs, reward, done, idx):
        bo_reward, cum_reward, battle_reward = self.update_fake_reward(next_obs)
        agent_obs = self._observation
        teacher_obs = {
            'spatial_info': agent_obs['spatial_info'],
            'entity_info': agent_obs['entity_info'],
            'scalar_info': agent_obs['scalar_info'],
            'entity_num': agent_obs['entity_num'],
            'selected_units_num': self._output['selected_units_num'],
            'action_info': self._output['action_info'],
        }
        server = getattr(self, '_batch_server', None)
        if server is not None and server.teacher_model is not None:
            teacher_output = self._batched_teacher_logit(teacher_obs)
        else:
            teacher_input = default_collate_with_dim([teacher_obs])
            teacher_input['hidden_state'] = [(h.unsqueeze(0), c.unsqueeze(0))
                                             for h, c in self._teacher_hidden_state]
            if self._use_cuda:
                teacher_input = to_device(teacher_input, 'cuda')
            teacher = self.teacher_model or self.model
            with torch.no_grad():
                teacher_output = teacher.compute_teacher_logit(**teacher_input)
            teacher_output = self.decollate_output(teacher_output)
        self._teacher_hidden_state = teacher_output['hidden_state']

        successive_output = None
        if self.successive_model is not None:
            succ_input = default_collate_with_dim([teacher_obs])
            succ_input['hidden_state'] = [(h.unsqueeze(0), c.unsqueeze(0))
                                          for h, c in self._successive_hidden_state]
            if self._use_cuda:
                succ_input = to_device(succ_input, 'cuda')
            with torch.no_grad():
                successive_output = self.successive_model.compute_teacher_logit(
                    **succ_input)
            successive_output = self.decollate_output(successive_output)
            self._successive_hidden_state = successive_output['hidden_state']

        action_info = copy.deepcopy(self._output['action_info'])
        at = int(action_info['action_type'])
        mask = {
            'actions_mask': {k: torch.tensor(v, dtype=torch.long)
                             for k, v in ACTIONS[at].items()
                             if k in ('queued', 'selected_units',
                                      'target_location', 'target_unit')},
            'cum_action_mask': torch.tensor(1.0),
            'build_order_mask': torch.tensor(float(self.use_bo_reward)),
            'built_unit_mask': torch.tensor(float(self.use_cum_reward)),
        }
        step_data = {
            'map_name': self._map_name,
            'spatial_info': agent_obs['spatial_info'],
            'model_last_iter': torch.tensor(self._model_last_iter, dtype=torch.float),
            'entity_info': agent_obs['entity_info'],
            'scalar_info': agent_obs['scalar_info'],
            'entity_num': agent_obs['entity_num'],
            'selected_units_num': self._output['selected_units_num'],
            'hidden_state': self._hidden_state_backup,
            'action_info': action_info,
            'behaviour_logp': self._output['action_logp'],
            'teacher_logit': teacher_output['logit'],
            'reward': {
                'winloss': torch.tensor(float(reward)),
                'build_order': bo_reward, 'built_unit': cum_reward,
                'battle': battle_reward,
            },
            'step': torch.tensor(float(self._game_step)),
            'mask': mask,
        }
        if successive_output is not None:
            step_data['successive_logit'] = successive_output['logit']
        if self._use_value_feature and 'value_feature' in agent_obs:
            step_data['value_feature'] = dict(agent_obs['value_feature'])
            step_data['value_feature'].update(self.get_behavior_z())
        self._hidden_state_backup = self._hidden_state
        self._data_buffer.append(step_data)
        self._push_count += 1
        if self._push_count == self._traj_len or done:
            last_obs = self._transform_obs(next_obs) if next_obs is not None \
                else dict(self._observation)
            last_obs.pop('game_info', None)
            last_step = {
                'map_name': self._map_name,
                'spatial_info': last_obs['spatial_info'],
                'entity_info': {k: v for k, v in last_obs['entity_info'].items()},
                'scalar_info': last_obs['scalar_info'],
                'entity_num': last_obs['entity_num'],
                'hidden_state': self._hidden_state,
            }
            if 'last_selected_units' not in last_step['entity_info']:
                width = last_step['entity_info']['x'].shape[0]
                last_step['entity_info']['last_selected_units'] = \
                    torch.zeros(width, dtype=torch.int8)
                last_step['entity_info']['last_targeted_unit'] = \
                    torch.zeros(width, dtype=torch.int8)
            if self._use_value_feature and 'value_feature' in last_obs:
                last_step['value_feature'] = dict(last_obs['value_feature'])
                last_step['value_feature'].update(self.get_behavior_z())
            # sliding window (reference agent.py:173,571-604): the deque is
            # NOT cleared between sends, so an episode-end window re-reaches
            # back to a full traj_len steps instead of going out short (the
            # learner collate requires uniform T)
            data = list(self._data_buffer) + [last_step]
            self._push_count = 0
            return data
        return None

    def get_behavior_z(self):
        bo = self._behaviour_building_order + \
            [0] * (BEGINNING_ORDER_LENGTH - len(self._behaviour_building_order))
        bo_loc = self._behaviour_bo_location + \
            [0] * (BEGINNING_ORDER_LENGTH - len(self._behaviour_bo_location))
        return {'beginning_order': torch.as_tensor(bo[:BEGINNING_ORDER_LENGTH], dtype=torch.long),
                'bo_location': torch.as_tensor(bo_loc[:BEGINNING_ORDER_LENGTH], dtype=torch.long),
                'cumulative_stat': torch.as_tensor(
                    self._behaviour_cumulative_stat, dtype=torch.bool).long()}

    # ---------------------------------------------------------- fake rewards
    def update_fake_reward(self, next_obs):
        return self._update_fake_reward(int(self._last_action_type),
                                        self._last_location, next_obs)

    def _update_fake_reward(self, action_type, location, next_obs):
        bo_reward = torch.zeros((), dtype=torch.float)
        cum_reward = torch.zeros((), dtype=torch.float)
        battle_reward = torch.zeros((), dtype=torch.float)
        if next_obs is None:
            return bo_reward, cum_reward, battle_reward
        # battle pseudo-reward: own score delta minus opponent score delta
        # (reference agent.py:623-626), computed even past the Z loop bound
        raw = next_obs.get('raw_obs') if isinstance(next_obs, dict) else None
        if raw is not None:
            from ..lib.features import compute_battle_score
            try:
                battle_score = compute_battle_score(raw)
                opp = next_obs.get('opponent_obs')
                opp_score = compute_battle_score(opp) if opp is not None \
                    else self._game_info['opponent_battle_score']
                battle_reward = torch.tensor(
                    (battle_score - self._game_info['battle_score']) -
                    (opp_score - self._game_info['opponent_battle_score']),
                    dtype=torch.float) / self._cfg['battle_norm']
            except AttributeError:      # mock obs without score protos
                pass
        if not self._exceed_flag:
            return bo_reward, cum_reward, battle_reward
        if action_type in BEGINNING_ORDER_ACTIONS and \
                (next_obs.get('action_result') or [1])[0] == 1:
            if action_type == 322:
                self._bo_zergling_count += 1
                if self._bo_zergling_count > 8:
                    return bo_reward, cum_reward, battle_reward
            order_index = BEGINNING_ORDER_ACTIONS.index(action_type)
            if order_index == 39 and 39 not in self._target_building_order:
                # spine crawler outside the target style: ignored (reference
                # agent.py:637-638)
                return bo_reward, cum_reward, battle_reward
            if len(self._behaviour_building_order) < len(self._target_building_order):
                self._behaviour_building_order.append(order_index)
                self._behaviour_bo_location.append(
                    int(location) if ACTIONS[action_type]['target_location'] else 0)
                if self.use_bo_reward:
                    if self._cfg['clip_bo']:    # compare against the target
                        tz = self._target_building_order[:len(self._behaviour_building_order)]
                        tz_lo = self._target_bo_location[:len(self._behaviour_building_order)]
                    else:
                        tz, tz_lo = self._target_building_order, self._target_bo_location
                    new_bo_dist = -levenshtein_distance(
                        torch.as_tensor(self._behaviour_building_order, dtype=torch.int),
                        tz.int(),
                        torch.as_tensor(self._behaviour_bo_location, dtype=torch.int),
                        tz_lo.int(),
                        partial(l2_distance, spatial_x=SPATIAL_SIZE[1])) / self._bo_norm
                    bo_reward = new_bo_dist - self._old_bo_reward
                    self._old_bo_reward = new_bo_dist
        cum_flag = False
        cum_type = self._cfg['cum_type']
        if cum_type == 'observation':
            # scan completed own units/upgrades (reference agent.py:663-677);
            # the first base at the born location is not a built structure
            cum_flag = True
            try:
                for u in next_obs['raw_obs'].observation.raw_data.units:
                    if u.alliance == 1 and u.unit_type in (59, 18, 86) and \
                            self._born_location_xy is not None and \
                            u.pos.x == self._born_location_xy[0] and \
                            u.pos.y == self._born_location_xy[1]:
                        continue
                    if u.alliance == 1 and u.build_progress == 1 and \
                            UNIT_TO_CUM[u.unit_type] != -1:
                        self._behaviour_cumulative_stat[UNIT_TO_CUM[u.unit_type]] = 1
                for uid in next_obs['raw_obs'].observation.raw_data.player.upgrade_ids:
                    if UPGRADE_TO_CUM[uid] != -1:
                        self._behaviour_cumulative_stat[UPGRADE_TO_CUM[uid]] = 1
            except (AttributeError, KeyError, TypeError):
                cum_flag = False        # mock obs without raw protos
        elif cum_type == 'action':
            action_name = ACTIONS[action_type]['name']
            if action_name in ('Cancel_quick', 'Cancel_Last_quick') and \
                    self._output is not None:
                # cancelling an in-progress train/build refunds its cum slot
                # (reference agent.py:682-696)
                try:
                    unit_index = int(self._output['action_info']['selected_units'][0])
                    order_len = int(self._observation['entity_info']['order_length'][unit_index])
                    action_index = None
                    if order_len == 1:
                        action_index = UNIT_ABILITY_TO_ACTION.get(
                            int(self._observation['entity_info']['order_id_0'][unit_index]))
                    elif order_len > 1:
                        qi = int(self._observation['entity_info']
                                 [f'order_id_{order_len - 1}'][unit_index]) - 1
                        if 0 <= qi < len(QUEUE_ACTIONS):
                            action_index = QUEUE_ACTIONS[qi]
                    if action_index in CUMULATIVE_STAT_ACTIONS:
                        cum_flag = True
                        ci = CUMULATIVE_STAT_ACTIONS.index(action_index)
                        self._behaviour_cumulative_stat[ci] = \
                            max(0, self._behaviour_cumulative_stat[ci] - 1)
                except (KeyError, IndexError, TypeError):
                    pass
            if action_type in CUMULATIVE_STAT_ACTIONS:
                cum_flag = True
                ci = CUMULATIVE_STAT_ACTIONS.index(action_type)
                self._behaviour_cumulative_stat[ci] += 1
        if self.use_cum_reward and cum_flag and \
                (cum_type == 'observation' or
                 (next_obs.get('action_result') or [1])[0] == 1):
            new_cum = -hamming_distance(
                torch.as_tensor(self._behaviour_cumulative_stat, dtype=torch.bool),
                self._target_cumulative_stat.bool()) / self._cum_norm
            cum_reward = (new_cum - self._old_cum_reward) * \
                self._get_time_factor(self._game_step)
            self._old_cum_reward = new_cum
        self._total_bo_reward += bo_reward
        self._total_cum_reward += cum_reward
        return bo_reward, cum_reward, battle_reward

    @staticmethod
    def _get_time_factor(game_step):
        """Cumulative-stat rewards fade late game (reference agent.py)."""
        if game_step < 10000:
            return 1.0
        if game_step < 20000:
            return 0.5
        return 0.25

    # ------------------------------------------------------------- telemetry
    def get_unit_num_info(self):
        return {'unit_num': self._stat_api.unit_num}

    def get_stat_data(self):
        data = self._stat_api.get_stat_data()
        bo_dist = levenshtein_distance(
            torch.as_tensor(self._behaviour_building_order, dtype=torch.int),
            self._target_building_order.int()).item()
        bo_dist_loc = levenshtein_distance(
            torch.as_tensor(self._behaviour_building_order, dtype=torch.int),
            self._target_building_order.int(),
            torch.as_tensor(self._behaviour_bo_location, dtype=torch.int),
            self._target_bo_location.int(),
            partial(l2_distance, spatial_x=SPATIAL_SIZE[1])).item()
        stat = {
            'race_id': self.race, 'step': self._game_step,
            'dist/bo': bo_dist, 'dist/bo_location': bo_dist_loc - bo_dist,
            'dist/cum': hamming_distance(
                torch.as_tensor(self._behaviour_cumulative_stat, dtype=torch.bool),
                self._target_cumulative_stat.bool()).item(),
            'bo_reward': self._total_bo_reward.item(),
            'cum_reward': self._total_cum_reward.item(),
            'bo_len': len(self._behaviour_building_order),
        }
        z0 = 0 if self.use_bo_reward else 1
        z1 = 0 if self.use_cum_reward else 1
        if z0:
            for k in ('dist/bo', 'bo_reward', 'bo_len', 'dist/bo_location'):
                stat[k] = None
        if z1:
            for k in ('dist/cum', 'cum_reward'):
                stat[k] = None
        stat['z_type'] = 2 * z1 + z0
        data.update(stat)
        cum_in, cum_out = defaultdict(int), defaultdict(int)
        for i in range(len(self._behaviour_cumulative_stat)):
            if self.race not in cum_dict[i]['race']:
                continue
            name = cum_dict[i]['name']
            built = self._behaviour_cumulative_stat[i] >= 1
            if self._target_cumulative_stat[i] < 1e-3:
                cum_out['cum_out/' + name] = int(built)
            else:
                cum_in['cum_in/' + name] = int(built)
        data.update(cum_in)
        data.update(cum_out)
        return data

    def set_model_last_iter(self, iteration):
        self._model_last_iter = iteration
