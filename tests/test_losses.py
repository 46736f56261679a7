"""End-to-end SL/RL loss tests on synthetic batches."""
import torch

from distar_amd.lib.fake_data import fake_rl_learner_data, fake_sl_batch
from distar_amd.losses import ReinforcementLoss, SupervisedLoss
from distar_amd.models import Model
from distar_amd.utils.config import Config


def test_sl_loss_finite_and_backward():
    torch.manual_seed(0)
    m = Model(Config({'common': {'type': 'train'}}))
    B, T = 2, 3
    data = fake_sl_batch(batch_size=B, traj_len=T, entity_num=64)
    hidden = [(torch.zeros(B, 384), torch.zeros(B, 384)) for _ in range(3)]
    logits, infer_action, _ = m.sl_train(**data, hidden_state=hidden)
    loss = SupervisedLoss(Config({'learner': {}}))
    ld = loss.compute_loss(logits, data['action_info'], data['action_mask'],
                           data['selected_units_num'], data['entity_num'], infer_action)
    total = ld['total_loss']
    assert torch.isfinite(total), ld
    assert float(total) < 1e4
    total.backward()
    grads = [p.grad for p in m.parameters() if p.grad is not None]
    assert grads and all(torch.isfinite(g).all() for g in grads)


def test_rl_loss_finite_and_backward():
    torch.manual_seed(0)
    m = Model(Config({'common': {'type': 'train'},
                      'model': {'enable_baselines':
                                ['winloss', 'build_order', 'built_unit', 'battle']}}),
              use_value_network=True)
    data = fake_rl_learner_data(batch_size=2, unroll_len=4, entity_num_range=(48, 96))
    data.pop('model_last_iter')
    data.pop('aux_type')
    out = m.rl_learner_forward(**data)
    loss = ReinforcementLoss(Config({}), 'MP0')
    ld = loss.compute_loss(out)
    total = ld['total_loss']
    assert torch.isfinite(total), {k: v for k, v in ld.items()
                                   if not isinstance(v, torch.Tensor) and abs(v) > 100}
    assert abs(float(total)) < 1e4
    total.backward()
    grads = [p.grad for p in m.parameters() if p.grad is not None]
    assert grads and all(torch.isfinite(g).all() for g in grads)


def test_rl_value_pretrain_freezes_policy():
    torch.manual_seed(0)
    m = Model(Config({'common': {'type': 'train'},
                      'model': {'enable_baselines': ['winloss']}}),
              use_value_network=True)
    m.only_update_baseline = True
    data = fake_rl_learner_data(batch_size=2, unroll_len=3, entity_num_range=(32, 48),
                                seed=1)
    data.pop('model_last_iter')
    data.pop('aux_type')
    out = m.rl_learner_forward(**data)
    loss = ReinforcementLoss(Config({}), 'MP0')
    loss.only_update_value = True
    ld = loss.compute_loss(out)
    ld['total_loss'].backward()
    # value nets get gradients; policy heads get none (critic input detached)
    assert any(p.grad is not None and p.grad.abs().sum() > 0
               for p in m.value_networks.parameters())
    assert all(p.grad is None or p.grad.abs().sum() == 0
               for p in m.policy.parameters())


def test_rl_loss_with_value_features():
    """use_value_feature path: ValueEncoder consumes the opponent-side
    features and feeds the critics (reference rl_user_config default)."""
    torch.manual_seed(0)
    m = Model(Config({'common': {'type': 'train'},
                      'learner': {'use_value_feature': True},
                      'model': {'enable_baselines':
                                ['winloss', 'build_order', 'built_unit', 'battle']}}),
              use_value_network=True)
    from distar_amd.lib.fake_data import fake_rl_learner_data_fast
    data = fake_rl_learner_data_fast(2, 3, entity_num=64, value_feature=True)
    data.pop('model_last_iter')
    out = m.rl_learner_forward(**data)
    ld = ReinforcementLoss(Config({}), 'MP0').compute_loss(out)
    assert torch.isfinite(ld['total_loss'])
    ld['total_loss'].backward()
    assert any(p.grad is not None and torch.isfinite(p.grad).all()
               for p in m.value_encoder.parameters())


def test_rl_loss_with_dapo():
    """DAPO successive-model KL (reference rl_loss.py:165-172), enabled for
    MainPlayer learners."""
    torch.manual_seed(0)
    m = Model(Config({'common': {'type': 'train'},
                      'model': {'enable_baselines': ['winloss']}}),
              use_value_network=True)
    from distar_amd.lib.fake_data import fake_rl_learner_data_fast
    data = fake_rl_learner_data_fast(2, 3, entity_num=64)
    data.pop('model_last_iter')
    out = m.rl_learner_forward(**data)
    out['successive_logit'] = {k: v.detach().clone()
                               for k, v in out['target_logit'].items()}
    loss = ReinforcementLoss(Config({'use_dapo': True}), 'MP0')
    assert loss.use_dapo
    ld = loss.compute_loss(out)
    assert 'dapo/total' in ld and torch.isfinite(ld['total_loss'])
    ld['total_loss'].backward()
    # non-main players silently disable dapo
    loss2 = ReinforcementLoss(Config({'use_dapo': True}), 'EP0')
    assert not loss2.use_dapo
