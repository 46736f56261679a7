"""GPU (MI355X) tests: HIP kernel numerics vs fp32 eager references, and the
train step end-to-end on device."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_hip_ext_loads():
    from distar_amd.ops import hip_ext
    assert hip_ext.available(), 'in-tree _hip_ops.so must be present on GPU boxes'


def test_vtrace_scan_matches_eager():
    from distar_amd.ops import scans
    torch.manual_seed(0)
    T, B = 64, 512
    rhos = torch.rand(T, B, device='cuda')
    cs = torch.rand(T, B, device='cuda')
    r = torch.randn(T, B, device='cuda')
    v = torch.randn(T + 1, B, device='cuda')
    g = torch.rand(T, B, device='cuda')
    lam = torch.rand(T, B, device='cuda')
    out = scans.vtrace_scan(rhos, cs, r, v, g, lam)
    ref = scans._vtrace_scan_eager(rhos.cpu(), cs.cpu(), r.cpu(), v.cpu(),
                                   g.cpu(), lam.cpu())
    torch.testing.assert_close(out.cpu(), ref, rtol=1e-5, atol=1e-5)


def test_lambda_return_scan_matches_eager():
    from distar_amd.ops import scans
    torch.manual_seed(1)
    T, B = 64, 384
    r = torch.randn(T, B, device='cuda')
    g = torch.rand(T, B, device='cuda')
    v = torch.randn(T, B, device='cuda')
    lam = torch.rand(T, B, device='cuda')
    out = scans.lambda_return_scan(r, g, v, lam)
    ref = scans._lambda_return_scan_eager(r.cpu(), g.cpu(), v.cpu(), lam.cpu())
    torch.testing.assert_close(out.cpu(), ref, rtol=1e-5, atol=1e-5)


def test_sl_train_step_on_gpu():
    from distar_amd.lib.fake_data import fake_sl_batch_fast
    from distar_amd.losses import SupervisedLoss
    from distar_amd.models import Model
    from distar_amd.utils.config import Config
    from distar_amd.utils.data import to_device
    torch.manual_seed(0)
    model = Model(Config({'common': {'type': 'train'}})).cuda()
    data = to_device(fake_sl_batch_fast(batch_size=2, traj_len=4), 'cuda')
    hidden = [(torch.zeros(2, 384, device='cuda'),
               torch.zeros(2, 384, device='cuda')) for _ in range(3)]
    loss_fn = SupervisedLoss(Config({'learner': {}}))
    with torch.autocast('cuda', dtype=torch.bfloat16):
        logits, infer_action, _ = model.sl_train(
            spatial_info=data['spatial_info'], scalar_info=data['scalar_info'],
            entity_info=data['entity_info'], entity_num=data['entity_num'],
            selected_units_num=data['selected_units_num'],
            traj_lens=data['traj_lens'], hidden_state=hidden,
            action_info=data['action_info'])
        ld = loss_fn.compute_loss(logits, data['action_info'], data['action_mask'],
                                  data['selected_units_num'], data['entity_num'],
                                  infer_action)
    assert torch.isfinite(ld['total_loss'])
    ld['total_loss'].backward()
    assert all(torch.isfinite(p.grad).all() for p in model.parameters()
               if p.grad is not None)


def test_rl_train_step_on_gpu():
    from distar_amd.lib.fake_data import fake_rl_learner_data_fast
    from distar_amd.losses import ReinforcementLoss
    from distar_amd.models import Model
    from distar_amd.utils.config import Config
    from distar_amd.utils.data import to_device
    torch.manual_seed(0)
    model = Model(Config({'common': {'type': 'train'},
                          'model': {'enable_baselines':
                                    ['winloss', 'build_order', 'built_unit',
                                     'battle']}}),
                  use_value_network=True).cuda()
    data = fake_rl_learner_data_fast(2, 8, entity_num=128)
    data.pop('model_last_iter')
    data = to_device(data, 'cuda')
    loss_fn = ReinforcementLoss(Config({}), 'MP0')
    with torch.autocast('cuda', dtype=torch.bfloat16):
        out = model.rl_learner_forward(**data)
        ld = loss_fn.compute_loss(out)
    assert torch.isfinite(ld['total_loss'])
    ld['total_loss'].backward()
    assert all(torch.isfinite(p.grad).all() for p in model.parameters()
               if p.grad is not None)
