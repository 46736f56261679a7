"""Checkpoint-layout parity + forward-mode shape checks for the AlphaStar
model (reference: distar/agent/default/model/model.py; golden layout
extracted by tools/extract_reference_tables.py)."""
import json
import os

import pytest
import torch

from distar_amd.lib.consts import (MAX_DELAY, MAX_ENTITY_NUM,
                                   MAX_SELECTED_UNITS_NUM, SPATIAL_SIZE)
from distar_amd.lib.fake_data import fake_obs_step, fake_sl_batch
from distar_amd.models import Model
from distar_amd.utils.config import Config
from distar_amd.utils.data import default_collate_with_dim

GOLDEN = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                      'distar_amd', 'assets', 'ckpt_layout_golden.json')


@pytest.fixture(scope='module')
def golden():
    with open(GOLDEN) as f:
        return json.load(f)


def test_policy_state_dict_layout(golden):
    m = Model(Config({'common': {'type': 'train'}}))
    mine = {k: list(v.shape) for k, v in m.state_dict().items()}
    assert mine == golden['policy']


def test_value_state_dict_layout(golden):
    m = Model(Config({'common': {'type': 'train'},
                      'learner': {'use_value_feature': True}}),
              use_value_network=True)
    mine = {k: list(v.shape) for k, v in m.state_dict().items()}
    assert mine == golden['value']


def test_reference_format_checkpoint_roundtrip(tmp_path, golden):
    """A checkpoint saved with the reference's dict layout loads drop-in."""
    from distar_amd.utils.checkpoint import CheckpointHelper, CountVar
    m = Model(Config({'common': {'type': 'train'}}))
    helper = CheckpointHelper()
    path = str(tmp_path / 'ckpt.pth.tar')
    helper.save(path, m, last_iter=CountVar(42))
    ckpt = torch.load(path, map_location='cpu', weights_only=False)
    assert set(ckpt['model'].keys()) == set(golden['policy'].keys())
    assert ckpt['last_iter'] == 42
    m2 = Model(Config({'common': {'type': 'train'}}))
    it = CountVar(0)
    helper.load(path, m2, last_iter=it)
    assert it.val == 42
    for (k1, v1), (k2, v2) in zip(m.state_dict().items(), m2.state_dict().items()):
        assert k1 == k2 and torch.equal(v1, v2)


@pytest.fixture(scope='module')
def model():
    torch.manual_seed(0)
    return Model(Config({'common': {'type': 'train'}}))


def test_sl_train_shapes(model):
    B, T, EN = 2, 3, 64
    data = fake_sl_batch(batch_size=B, traj_len=T, entity_num=EN)
    hidden = [(torch.zeros(B, 384), torch.zeros(B, 384)) for _ in range(3)]
    logits, infer_action, out_state = model.sl_train(**data, hidden_state=hidden)
    n = B * T
    assert logits['action_type'].shape == (n, 327)
    assert logits['delay'].shape == (n, MAX_DELAY + 1)
    assert logits['queued'].shape == (n, 2)
    assert logits['selected_units'].shape[0] == n
    assert logits['selected_units'].shape[2] == MAX_ENTITY_NUM + 1
    assert logits['target_unit'].shape == (n, MAX_ENTITY_NUM)
    assert logits['target_location'].shape == (n, SPATIAL_SIZE[0] * SPATIAL_SIZE[1])
    assert len(out_state) == 3 and out_state[0][0].shape == (B, 384)


def test_inference_and_teacher_shapes(model):
    B = 2
    obs = default_collate_with_dim([fake_obs_step(entity_num=48) for _ in range(B)])
    hidden = [(torch.zeros(B, 384), torch.zeros(B, 384)) for _ in range(3)]
    with torch.no_grad():
        out = model.compute_logp_action(**obs, hidden_state=hidden)
    assert out['action_info']['action_type'].shape == (B,)
    assert out['action_info']['selected_units'].shape[0] == B
    assert (out['selected_units_num'] <= MAX_SELECTED_UNITS_NUM).all()
    for k in ('action_type', 'delay', 'queued', 'target_unit', 'target_location'):
        assert out['action_logp'][k].shape == (B,)
    with torch.no_grad():
        t = model.compute_teacher_logit(**obs, hidden_state=hidden,
                                        selected_units_num=out['selected_units_num'],
                                        action_info=out['action_info'])
    assert t['logit']['action_type'].shape == (B, 327)
    assert t['logit']['selected_units'].shape[0] == B


def test_selected_units_sampling_semantics(model):
    """Sampled selections: no duplicates before the end token; num matches
    the end-token position; end token never first."""
    torch.manual_seed(3)
    B = 4
    obs = default_collate_with_dim([fake_obs_step(entity_num=32) for _ in range(B)])
    hidden = [(torch.zeros(B, 384), torch.zeros(B, 384)) for _ in range(3)]
    with torch.no_grad():
        out = model.compute_logp_action(**obs, hidden_state=hidden)
    su = out['action_info']['selected_units']
    num = out['selected_units_num']
    en = obs['entity_num']
    for b in range(B):
        n = int(num[b])
        sel = su[b, :n].tolist()
        if n == 0:
            continue
        body = [s for s in sel if s != int(en[b])]
        assert len(body) == len(set(body)), 'duplicate selection'
        if n < su.shape[1]:
            assert sel[-1] == int(en[b]) or len(sel) == MAX_SELECTED_UNITS_NUM


def test_action_type_head_race_mask_play_mode():
    """In play mode the 327-way logits are masked to the agent race's legal
    actions (reference action_type_head.py:53-55 + stat.py ACTION_RACE_MASK)."""
    from distar_amd.lib.stat import ACTION_RACE_MASK
    from distar_amd.models.alphastar.heads import ActionTypeHead
    from distar_amd.utils.config import Config
    from distar_amd.models.alphastar.model import alphastar_model_default_config
    torch.manual_seed(0)
    cfg = Config({'common': {'type': 'play'}})
    from distar_amd.utils.config import deep_merge_dicts
    cfg = deep_merge_dicts(alphastar_model_default_config, cfg)
    head = ActionTypeHead(cfg)
    assert head.use_mask
    for race in ('zerg', 'terran', 'protoss'):
        head.race = race
        logits, action, _ = head(torch.randn(4, 384), torch.randn(4, 448))
        mask = ACTION_RACE_MASK[race]
        assert (logits[:, ~mask] <= -1e8).all()
        assert mask[action].all()          # sampled actions are race-legal
    # train mode leaves logits unmasked
    head2 = ActionTypeHead(alphastar_model_default_config)
    assert not head2.use_mask
