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
        if 'successive_logit' in traj_data:     # DAPO lagged-self logits
            sl = traj_data['successive_logit']
            s_su_pad = MAX_SELECTED_UNITS_NUM - sl['selected_units'].shape[0]
            sl['selected_units'] = torch.nn.functional.pad(
                sl['selected_units'], (0, pad, 0, s_su_pad), 'constant', -1e9)
            sl['target_unit'] = torch.nn.functional.pad(
                sl['target_unit'], (0, pad), 'constant', -1e9)
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


def _batched_obs_tensors(n, entity_width, seed):
    """Vectorized synthetic obs: batched tensors built directly (no per-step
    Python loop) — same shapes/dtypes/vocabularies as fake_step_data."""
    from .consts import SPATIAL_INFO, SCALAR_INFO, ENTITY_INFO, EFFECT_LEN
    g = torch.Generator().manual_seed(seed)

    def ri(high, size, dtype):
        return torch.randint(0, max(int(high), 1), size=size, dtype=dtype, generator=g)

    spatial_info = {}
    for k, dtype in SPATIAL_INFO:
        if 'effect' in k:
            spatial_info[k] = ri(SPATIAL_SIZE[0] * SPATIAL_SIZE[1], (n, EFFECT_LEN), dtype)
        else:
            high = {'height_map': 256, 'visibility_map': 4, 'player_relative': 5}.get(k, 2)
            spatial_info[k] = ri(high, (n, *SPATIAL_SIZE), dtype)
    scalar_info = {}
    for k, dtype, size in SCALAR_INFO:
        if k == 'time':
            scalar_info[k] = torch.rand((n,), generator=g) * 1000
        elif k == 'agent_statistics':
            scalar_info[k] = torch.rand((n, *size), generator=g) * 10
        elif k == 'beginning_order':
            scalar_info[k] = ri(174, (n, *size), dtype)
        else:
            high = {'home_race': 5, 'away_race': 5, 'last_queued': 2,
                    'last_delay': MAX_DELAY + 1, 'last_action_type': NUM_ACTIONS,
                    'bo_location': SPATIAL_SIZE[0] * SPATIAL_SIZE[1]}.get(k, 2)
            scalar_info[k] = ri(high, (n, *size), dtype)
    entity_info = {}
    ENT_HIGH = {'unit_type': 260, 'alliance': 5, 'cargo_space_taken': 9,
                'display_type': 5, 'x': SPATIAL_SIZE[1], 'y': SPATIAL_SIZE[0],
                'cloak': 5, 'cargo_space_max': 9, 'assigned_harvesters': 24,
                'weapon_cooldown': 32, 'order_length': 9, 'order_id_0': NUM_ACTIONS,
                'order_id_1': 49, 'buff_id_0': 50, 'buff_id_1': 50,
                'addon_unit_type': 9, 'order_id_2': 49, 'order_id_3': 49,
                'attack_upgrade_level': 4, 'armor_upgrade_level': 4,
                'shield_upgrade_level': 4}
    for k, dtype in ENTITY_INFO:
        if dtype in (torch.float16, torch.float32):
            entity_info[k] = torch.rand((n, entity_width), generator=g).to(dtype)
        else:
            entity_info[k] = ri(ENT_HIGH.get(k, 2), (n, entity_width), dtype)
    return spatial_info, scalar_info, entity_info, g


def fake_sl_batch_fast(batch_size=32, traj_len=64, entity_min=100, seed=0):
    """Vectorized SL batch at the reference SL layout: fixed 512-entity slab
    width (the reference's shared-memory slab always runs the transformer on
    512 entities with a mask), (B*T) lane-major rows."""
    n = batch_size * traj_len
    spatial_info, scalar_info, entity_info, g = _batched_obs_tensors(
        n, MAX_ENTITY_NUM, seed)
    entity_num = torch.randint(entity_min, MAX_ENTITY_NUM, (n,), generator=g)
    su_num = torch.minimum(
        torch.randint(1, MAX_SELECTED_UNITS_NUM, (n,), generator=g), entity_num)
    # distinct selections per row (argsort of masked rand restricts the
    # permutation to each row's valid entities), end token at su_num-1
    r = torch.rand(n, MAX_ENTITY_NUM, generator=g)
    r[torch.arange(MAX_ENTITY_NUM).unsqueeze(0) >= entity_num.unsqueeze(1)] = 2.0
    sel = r.argsort(dim=1)[:, :MAX_SELECTED_UNITS_NUM]
    sel.scatter_(1, (su_num - 1).unsqueeze(1), entity_num.unsqueeze(1))
    action_info = {
        'action_type': torch.randint(0, NUM_ACTIONS, (n,), generator=g),
        'delay': torch.randint(0, MAX_DELAY + 1, (n,), generator=g),
        'queued': torch.randint(0, 2, (n,), generator=g),
        'selected_units': sel.long(),
        'target_unit': (torch.randint(0, 1 << 30, (n,), generator=g) % entity_num).long(),
        'target_location': torch.randint(0, SPATIAL_SIZE[0] * SPATIAL_SIZE[1], (n,),
                                         generator=g),
    }
    action_mask = {k: torch.ones(n, dtype=torch.bool) for k in ACTION_INFO}
    return {
        'spatial_info': spatial_info, 'scalar_info': scalar_info,
        'entity_info': entity_info, 'entity_num': entity_num,
        'action_info': action_info, 'action_mask': action_mask,
        'selected_units_num': su_num,
        'traj_lens': [traj_len] * batch_size,
        'new_episodes': [False] * batch_size,
    }


def fake_value_feature(n, entity_num, g):
    """Synthetic opponent-side value features (reference
    `lib/features.py:735-765` value_feature dict), batched (n, ...)."""
    from .actions import NUM_CUMULATIVE_STAT_ACTIONS
    H, W = SPATIAL_SIZE
    return {
        'enemy_unit_counts_bow': torch.randint(0, 3, (n, 260), generator=g).to(torch.uint8),
        'enemy_unit_type_bool': torch.randint(0, 2, (n, 260), generator=g).to(torch.uint8),
        'enemy_agent_statistics': torch.rand(n, 10, generator=g) * 5,
        'enemy_upgrades': torch.randint(0, 2, (n, 90), generator=g).to(torch.uint8),
        'unit_alliance': torch.randint(0, 2, (n, MAX_ENTITY_NUM), generator=g),
        'unit_type': torch.randint(0, 260, (n, MAX_ENTITY_NUM), generator=g),
        'beginning_order': torch.randint(0, 174, (n, 20), generator=g),
        'bo_location': torch.randint(0, H * W, (n, 20), generator=g),
        'cumulative_stat': torch.randint(0, 2, (n, NUM_CUMULATIVE_STAT_ACTIONS),
                                         generator=g),
        'unit_x': torch.randint(0, W, (n, MAX_ENTITY_NUM), generator=g),
        'unit_y': torch.randint(0, H, (n, MAX_ENTITY_NUM), generator=g),
        'total_unit_count': torch.randint(entity_num // 2, MAX_ENTITY_NUM, (n,),
                                          generator=g),
        'own_units_spatial': torch.randint(0, 2, (n, 1, H, W), generator=g).bool(),
        'enemy_units_spatial': torch.randint(0, 2, (n, 1, H, W), generator=g).bool(),
    }


def fake_rl_learner_data_fast(batch_size=4, unroll_len=16, entity_num=256, seed=0,
                              value_feature=False):
    """Vectorized RL learner batch (uniform entity width = batch-max padding
    outcome), time-major obs over (T+1)*B rows like the RL collate."""
    T, B, EN = unroll_len, batch_size, entity_num
    n = (T + 1) * B
    spatial_info, scalar_info, entity_info, g = _batched_obs_tensors(n, EN, seed)
    entity_nums = torch.randint(EN // 2, EN, (n,), generator=g)
    su_num = torch.minimum(
        torch.randint(1, MAX_SELECTED_UNITS_NUM, (T, B), generator=g),
        entity_nums[:T * B].view(T, B))
    r = torch.rand(T * B, EN, generator=g)
    en_flat = entity_nums[:T * B]
    r[torch.arange(EN).unsqueeze(0) >= en_flat.unsqueeze(1)] = 2.0
    sel = r.argsort(dim=1)[:, :MAX_SELECTED_UNITS_NUM]
    if sel.shape[1] < MAX_SELECTED_UNITS_NUM:
        sel = torch.nn.functional.pad(sel, (0, MAX_SELECTED_UNITS_NUM - sel.shape[1]))
    sel.scatter_(1, (su_num.view(-1) - 1).unsqueeze(1), en_flat.unsqueeze(1))
    sel = sel.view(T, B, MAX_SELECTED_UNITS_NUM).long()
    action_info = {
        'action_type': torch.randint(0, NUM_ACTIONS, (T, B), generator=g),
        'delay': torch.randint(0, MAX_DELAY + 1, (T, B), generator=g),
        'queued': torch.randint(0, 2, (T, B), generator=g),
        'selected_units': sel,
        'target_unit': (torch.randint(0, 1 << 30, (T, B), generator=g)
                        % en_flat.view(T, B)).long(),
        'target_location': torch.randint(0, SPATIAL_SIZE[0] * SPATIAL_SIZE[1], (T, B),
                                         generator=g),
    }
    su_mask = sequence_mask(su_num.view(-1), max_len=MAX_SELECTED_UNITS_NUM).view(
        T, B, MAX_SELECTED_UNITS_NUM)
    behaviour_logp = {
        'action_type': -torch.rand(T, B), 'delay': -torch.rand(T, B),
        'queued': -torch.rand(T, B),
        'selected_units': torch.where(su_mask, -torch.rand(T, B, MAX_SELECTED_UNITS_NUM),
                                      torch.full((T, B, MAX_SELECTED_UNITS_NUM), -1e9)),
        'target_unit': -torch.rand(T, B), 'target_location': -torch.rand(T, B),
    }
    # teacher SU logits with the availability/prev-selected mask structure
    su_logit = torch.randn(T, B, MAX_SELECTED_UNITS_NUM, EN + 1, generator=g)
    avail = sequence_mask(en_flat + 1, max_len=EN + 1).view(T, B, 1, EN + 1)
    step_idx = torch.arange(MAX_SELECTED_UNITS_NUM).view(1, 1, -1, 1)
    valid_step = su_mask.unsqueeze(-1)
    # previously-selected mask: labels before step s disabled
    onehot = torch.zeros(T * B, MAX_SELECTED_UNITS_NUM, EN + 1, dtype=torch.int16)
    onehot.scatter_(2, sel.view(T * B, -1, 1), 1)
    prev = torch.cat([onehot.new_zeros(T * B, 1, EN + 1),
                      onehot.cumsum(1)[:, :-1]], dim=1) > 0
    mask_full = avail & ~prev.view(T, B, MAX_SELECTED_UNITS_NUM, EN + 1) & valid_step
    mask_full[:, :, 0].scatter_(-1, en_flat.view(T, B, 1), False)
    su_logit = su_logit.masked_fill(~mask_full, -1e9)
    tu_logit = torch.randn(T, B, EN, generator=g).masked_fill(
        ~sequence_mask(en_flat, max_len=EN).view(T, B, EN), -1e9)
    teacher_logit = {
        'action_type': torch.randn(T, B, NUM_ACTIONS, generator=g),
        'delay': torch.randn(T, B, MAX_DELAY + 1, generator=g),
        'queued': torch.randn(T, B, 2, generator=g),
        'selected_units': su_logit, 'target_unit': tu_logit,
        'target_location': torch.randn(T, B, SPATIAL_SIZE[0] * SPATIAL_SIZE[1],
                                       generator=g),
    }
    mask = {
        'actions_mask': {k: torch.ones(T, B, dtype=torch.long)
                         for k in ('queued', 'selected_units', 'target_location',
                                   'target_unit')},
        'cum_action_mask': torch.ones(T, B), 'build_order_mask': torch.ones(T, B),
        'built_unit_mask': torch.ones(T, B),
        'selected_units_mask': su_mask,
        'selected_units_logits_mask': sequence_mask(en_flat + 1, max_len=EN + 1
                                                    ).view(T, B, EN + 1),
        'target_units_logits_mask': sequence_mask(en_flat, max_len=EN).view(T, B, EN),
    }
    reward = {
        'winloss': torch.randint(-1, 2, (T, B), generator=g).float(),
        'build_order': torch.rand(T, B, generator=g) * 2 - 1,
        'built_unit': torch.rand(T, B, generator=g) * 2 - 1,
        'battle': torch.rand(T, B, generator=g) * 2 - 1,
    }
    out = {
        'spatial_info': spatial_info, 'scalar_info': scalar_info,
        'entity_info': entity_info, 'entity_num': entity_nums,
        'hidden_state': [(torch.zeros(n, 384), torch.zeros(n, 384)) for _ in range(3)],
        'action_info': action_info, 'selected_units_num': su_num,
        'behaviour_logp': behaviour_logp, 'teacher_logit': teacher_logit,
        'mask': mask, 'reward': reward,
        'step': torch.randint(0, 10000, (T, B), generator=g).float(),
        'model_last_iter': torch.zeros(B),
        'batch_size': B, 'unroll_len': T,
    }
    if value_feature:
        out['value_feature'] = fake_value_feature(n, EN, g)
    return out


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
