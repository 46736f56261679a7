"""Synthetic trajectory builders for tests and benchmarks.

The per-step dict layout matches what the actor's `collect_data` produces
(reference `agent/default/agent.py:475-607`), and the RL collate mirrors the
reference's `rl_training/rl_dataloader.py:collate_fn/padding_entity_info`
(batch-max entity padding, SU/teacher-logit -1e9 padding, (T+1) obs frames vs
T action frames, time-major flatten).

BASELINE.json requires the headline metrics measured on synthetic data of the
reference's shapes — these builders are that data source.
"""
import random

import torch

from .actions import ACTIONS, NUM_ACTIONS
from .consts import (ACTION_INFO, MAX_DELAY, MAX_ENTITY_NUM,
                     MAX_SELECTED_UNITS_NUM, SPATIAL_SIZE, fake_step_data)
from ..models.nn.blocks import sequence_mask
from ..utils.data import default_collate_with_dim, flat

RL_ENABLED_BASELINES = ['winloss', 'build_order', 'built_unit', 'battle']


def fake_obs_step(entity_num=None, randomize=True):
    return fake_step_data(train=False, entity_num=entity_num, randomize=randomize)


def fake_selected_units(entity_num, su_num, pad_to=None):
    """Distinct entity indices with the end-token (index entity_num) last,
    like real decoded selections.  Padded with 0 to ``pad_to`` if given."""
    sel = torch.randperm(entity_num)[:su_num - 1]
    sel = torch.cat([sel, torch.tensor([entity_num])])
    if pad_to is not None and sel.shape[0] < pad_to:
        sel = torch.nn.functional.pad(sel, (0, pad_to - sel.shape[0]), 'constant', 0)
    return sel.long()


def _masked_su_teacher_logit(labels, entity_num, su_num):
    """Random teacher logits carrying the SAME mask structure a real teacher
    forward produces (availability, minus previously-selected, end flag off at
    step 0) — real teacher logits are masked identically, and the KL term
    relies on it (teacher-valid/target-masked positions would blow up)."""
    logit = torch.randn(su_num, entity_num + 1)
    logit[0, entity_num] = -1e9
    for s in range(1, su_num):
        logit[s, labels[:s]] = -1e9
    return logit


def fake_rl_step(entity_num, rng: random.Random, randomize=True):
    """One actor step: obs + action + behaviour logp + teacher logit + reward
    + masks, shaped like `collect_data`'s step_data."""
    step = fake_step_data(train=True, entity_num=entity_num, randomize=randomize)
    su_num = min(rng.randint(1, MAX_SELECTED_UNITS_NUM), entity_num)
    step['selected_units_num'] = torch.tensor(su_num, dtype=torch.long)
    action_info = step['action_info']
    action_info['selected_units'] = fake_selected_units(entity_num, su_num)
    action_info['target_unit'] = torch.randint(0, entity_num, (), dtype=torch.long)
    at = int(action_info['action_type'])
    flags = {k: v for k, v in ACTIONS[at].items()
             if k not in ('name', 'goal', 'func_id', 'general_ability_id', 'game_id')}
    mask = {
        'actions_mask': {k: torch.tensor(v, dtype=torch.long) for k, v in flags.items()},
        'cum_action_mask': torch.tensor(1.0),
        'build_order_mask': torch.tensor(1.0),
        'built_unit_mask': torch.tensor(1.0),
    }
    step.pop('action_mask')
    step.update({
        'model_last_iter': torch.tensor(0, dtype=torch.float),
        'hidden_state': [(torch.zeros(384), torch.zeros(384)) for _ in range(3)],
        'behaviour_logp': {
            'action_type': torch.tensor(-1.0), 'delay': torch.tensor(-1.0),
            'queued': torch.tensor(-1.0),
            'selected_units': torch.full((su_num,), -1.0),
            'target_unit': torch.tensor(-1.0), 'target_location': torch.tensor(-1.0),
        },
        'teacher_logit': {
            'action_type': torch.randn(NUM_ACTIONS), 'delay': torch.randn(MAX_DELAY + 1),
            'queued': torch.randn(2),
            'selected_units': _masked_su_teacher_logit(
                action_info['selected_units'], entity_num, su_num),
            'target_unit': torch.randn(entity_num),
            'target_location': torch.randn(SPATIAL_SIZE[0] * SPATIAL_SIZE[1]),
        },
        'reward': {
            'winloss': torch.tensor(float(rng.choice([-1, 0, 1]))),
            'build_order': torch.tensor(rng.uniform(-1, 1)),
            'built_unit': torch.tensor(rng.uniform(-1, 1)),
            'battle': torch.tensor(rng.uniform(-1, 1)),
        },
        'step': torch.tensor(float(rng.randint(0, 10000))),
        'mask': mask,
    })
    # trim obs fields to the real entity_num (collate re-pads to batch max)
    for k in step['entity_info']:
        step['entity_info'][k] = step['entity_info'][k][:entity_num]
    action_info['selected_units'] = action_info['selected_units'][:su_num]
    return step


def padding_entity_info(traj_data, max_entity_num):
    """Reference `rl_dataloader.py:padding_entity_info`."""
    traj_data.pop('map_name', None)
    pad = max_entity_num - len(traj_data['entity_info']['x'])
    for k in traj_data['entity_info']:
        traj_data['entity_info'][k] = torch.nn.functional.pad(
            traj_data['entity_info'][k], (0, pad), 'constant', 0)
    if 'action_info' in traj_data:
        su_pad = MAX_SELECTED_UNITS_NUM - traj_data['teacher_logit']['selected_units'].shape[0]
        traj_data['mask']['selected_units_mask'] = sequence_mask(
            traj_data['selected_units_num'].unsqueeze(0),
            max_len=MAX_SELECTED_UNITS_NUM).squeeze(0)
        traj_data['action_info']['selected_units'] = torch.nn.functional.pad(
            traj_data['action_info']['selected_units'],
            (0, MAX_SELECTED_UNITS_NUM - traj_data['action_info']['selected_units'].shape[-1]),
            'constant', 0)
        traj_data['behaviour_logp']['selected_units'] = torch.nn.functional.pad(
            traj_data['behaviour_logp']['selected_units'], (0, su_pad), 'constant', -1e9)
        traj_data['teacher_logit']['selected_units'] = torch.nn.functional.pad(
            traj_data['teacher_logit']['selected_units'], (0, pad, 0, su_pad), 'constant', -1e9)
        traj_data['teacher_logit']['target_unit'] = torch.nn.functional.pad(
            traj_data['teacher_logit']['target_unit'], (0, pad), 'constant', -1e9)
        traj_data['mask']['selected_units_logits_mask'] = sequence_mask(
            traj_data['entity_num'].unsqueeze(0) + 1, max_len=max_entity_num + 1).squeeze(0)
        traj_data['mask']['target_units_logits_mask'] = sequence_mask(
            traj_data['entity_num'].unsqueeze(0), max_len=max_entity_num).squeeze(0)
    return traj_data


def rl_collate(traj_batch):
    """Reference `rl_dataloader.py:collate_fn`: per-traj stack (dim 0 = time),
    then batch stack (dim 1), obs keys flattened time-major."""
    max_entity_num = max(len(td['entity_info']['x'])
                         for traj in traj_batch for td in traj)
    traj_batch = [[padding_entity_info(td, max_entity_num) for td in traj]
                  for traj in traj_batch]
    data = [default_collate_with_dim(traj) for traj in traj_batch]
    batch_size = len(data)
    unroll_len = len(data[0]['step'])
    data = default_collate_with_dim(data, dim=1)
    new_data = {}
    for k, val in data.items():
        if k in ('spatial_info', 'entity_info', 'scalar_info', 'entity_num',
                 'entity_location', 'hidden_state', 'value_feature'):
            new_data[k] = flat(val)
        else:
            new_data[k] = val
    new_data['aux_type'] = batch_size
    new_data['batch_size'] = batch_size
    new_data['unroll_len'] = unroll_len
    return new_data


def fake_rl_learner_data(batch_size=4, unroll_len=16, entity_num_range=(64, 256),
                         seed=0):
    """Collated RL learner batch: T action frames + 1 bootstrap obs frame per
    lane, ready for `Model.rl_learner_forward(**data)` (after popping
    'model_last_iter', 'aux_type')."""
    rng = random.Random(seed)
    torch.manual_seed(seed)
    trajs = []
    for _ in range(batch_size):
        en = rng.randint(*entity_num_range)
        steps = [fake_rl_step(en, rng) for _ in range(unroll_len)]
        last = fake_obs_step(entity_num=en)
        last['hidden_state'] = [(torch.zeros(384), torch.zeros(384)) for _ in range(3)]
        for k in last['entity_info']:
            last['entity_info'][k] = last['entity_info'][k][:en]
        steps.append(last)
        trajs.append(steps)
    return rl_collate(trajs)


def fake_sl_batch(batch_size=4, traj_len=8, entity_num=None, seed=0):
    """SL batch: (B*T) lane-major rows + traj_lens/new_episodes
    (reference `sl_dataloader.py:__next__` / FakeDataloader)."""
    torch.manual_seed(seed)
    rng = random.Random(seed)
    steps = []
    for _ in range(batch_size * traj_len):
        en = entity_num or rng.randint(64, MAX_ENTITY_NUM)
        s = fake_step_data(train=True, entity_num=en, randomize=True)
        su = max(min(int(s['selected_units_num']), en), 1)
        s['selected_units_num'] = torch.tensor(su, dtype=torch.long)
        s['action_info']['selected_units'] = fake_selected_units(en, su,
                                                                 pad_to=MAX_SELECTED_UNITS_NUM)
        steps.append(s)
    data = default_collate_with_dim(steps)
    data['traj_lens'] = [traj_len] * batch_size
    data['new_episodes'] = [False] * batch_size
    return data
