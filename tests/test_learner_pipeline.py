"""Learner-loop tests: SL learner end-to-end with the synthetic dataloader,
hook wiring, checkpoint save/load resume, and the offline SL dataloader's
shared-memory lane protocol."""
import glob
import os

import pytest
import torch

from distar_amd.learner.sl_learner import SLLearner
from distar_amd.utils.config import Config


def _sl_cfg(tmp_path, **learner_overrides):
    learner = {
        'job_type': 'fake', 'use_cuda': False, 'use_amp': False,
        'learning_rate': 1e-4,
        'data': {'batch_size': 2, 'trajectory_length': 3},
        'hook': {'after_iter': {'log_show': {'ext_args': {'freq': 1000}},
                                'save_ckpt': {'ext_args': {'freq': 2}}}},
    }
    learner.update(learner_overrides)
    return Config({'common': {'experiment_name': 'test_sl',
                              'experiment_dir': str(tmp_path), 'type': 'train'},
                   'learner': learner})


@pytest.mark.timeout(900)
def test_sl_learner_runs_and_checkpoints(tmp_path):
    torch.manual_seed(0)
    learner = SLLearner(_sl_cfg(tmp_path))
    learner.run(max_iterations=2)
    assert learner.last_iter.val == 2
    assert 'total_loss' in learner.record.var_dict
    ckpts = glob.glob(str(tmp_path / 'test_sl' / 'checkpoint' / '*.pth.tar'))
    assert ckpts, 'save_ckpt hook did not fire'
    # resume: new learner loads the checkpoint and continues
    cfg = _sl_cfg(tmp_path, load_path=ckpts[-1])
    learner2 = SLLearner(cfg)
    learner2.call_hook('before_run')
    assert learner2.last_iter.val == 2


@pytest.mark.timeout(900)
def test_offline_sl_dataloader_lanes(tmp_path):
    """Shared-memory lane protocol with the 'offline' source: pre-decoded
    step files stream through worker processes into the shared batch."""
    from distar_amd.data.sl_dataloader import SLDataloader
    from distar_amd.lib.consts import fake_step_data
    torch.manual_seed(0)
    data_dir = tmp_path / 'decoded'
    data_dir.mkdir()
    for i in range(2):
        steps = [fake_step_data(train=True, entity_num=32, randomize=True)
                 for _ in range(5)]
        torch.save(steps, data_dir / f'replay_{i}.pt')
    cfg = Config({'learner': {
        'use_cuda': False, 'use_distributed': False,
        'data': {'source': 'offline', 'train_data_file': str(data_dir),
                 'batch_size': 2, 'trajectory_length': 3, 'num_workers': 2,
                 'epochs': 100}}})
    loader = SLDataloader(cfg)
    try:
        batch1 = next(loader)
        assert batch1['traj_lens'] == [3, 3]
        assert all(batch1['new_episodes'])
        assert batch1['entity_num'].shape[0] == 6     # B*T rows
        batch2 = next(loader)
        # second window of a 5-step replay: 2 valid steps
        assert batch2['traj_lens'] == [2, 2]
        assert not any(batch2['new_episodes'])
    finally:
        loader.close()


@pytest.mark.timeout(900)
def test_auto_checkpoint_on_crash(tmp_path):
    """auto_checkpoint saves an emergency checkpoint when the train loop
    raises (reference checkpoint_helper.py:325-369)."""

    class Boom(RuntimeError):
        pass

    class CrashingLearner(SLLearner):
        def _train(self, data):
            if self.last_iter.val >= 1:
                raise Boom('injected fault')
            super()._train(data)

    learner = CrashingLearner(_sl_cfg(tmp_path))
    with pytest.raises(Boom):
        learner.run(max_iterations=5)
    ckpts = glob.glob(str(tmp_path / 'test_sl' / 'checkpoint' / '*.pth.tar'))
    assert ckpts, 'no emergency checkpoint saved'


@pytest.mark.timeout(900)
def test_sl_learner_learns_fixed_batch(tmp_path):
    """Trainability: overfitting one fixed synthetic batch reduces the SL
    loss substantially (catches sign errors / dead gradients that shape
    tests cannot)."""
    import torch
    from distar_amd.lib.fake_data import fake_sl_batch
    from distar_amd.losses import SupervisedLoss
    from distar_amd.models import Model
    from distar_amd.utils.config import Config
    torch.manual_seed(0)
    model = Model(Config({'common': {'type': 'train'}}))
    data = fake_sl_batch(batch_size=2, traj_len=2, entity_num=32)
    hidden = [(torch.zeros(2, 384), torch.zeros(2, 384)) for _ in range(3)]
    loss_fn = SupervisedLoss(Config({'learner': {}}))
    opt = torch.optim.Adam(model.parameters(), lr=3e-4)
    losses = []
    for _ in range(12):
        logits, infer_action, _ = model.sl_train(**data, hidden_state=hidden)
        ld = loss_fn.compute_loss(logits, data['action_info'], data['action_mask'],
                                  data['selected_units_num'], data['entity_num'],
                                  infer_action)
        opt.zero_grad()
        ld['total_loss'].backward()
        opt.step()
        losses.append(float(ld['total_loss'].detach()))
    assert losses[-1] < losses[0] * 0.5, losses


@pytest.mark.timeout(300)
def test_rl_learner_debug_endpoints(tmp_path, monkeypatch):
    """Live-control HTTP endpoints (reference rl_learner.py:263-287):
    update_config rebuilds the loss with merged overrides, reset_value
    reinitializes value networks at the next iteration."""
    monkeypatch.chdir(tmp_path)
    import requests
    from distar_amd.learner.rl_learner import RLLearner
    torch.manual_seed(0)
    cfg = Config({'learner': {'player_id': 'MP0', 'job_type': 'fake',
                              'use_cuda': False, 'use_amp': False,
                              'data': {'batch_size': 2, 'trajectory_length': 3},
                              'hook': {'after_iter': {
                                  'log_show': {'ext_args': {'freq': 1000}}}}},
                  'common': {'experiment_name': 'test_debug_ep', 'type': 'train'},
                  'model': {'enable_baselines': ['winloss']}})
    learner = RLLearner(cfg)
    srv = learner.start_debug_server()
    url = f'http://127.0.0.1:{srv.port}'
    old_weight = float(learner._loss.loss_weights.kl)
    r = requests.post(f'{url}/learner/update_config', json={
        'overrides': {'learner': {'loss_weights': {'kl': old_weight * 2}}}})
    assert r.json().get('done')
    r = requests.post(f'{url}/learner/reset_value', json={})
    assert r.json().get('done')
    learner.run(max_iterations=1)
    assert float(learner._loss.loss_weights.kl) == old_weight * 2
    assert not learner._reset_value_flag
    srv.stop()
    learner._dataloader.close()


@pytest.mark.timeout(300)
def test_remote_sl_dataloader_via_adapter(tmp_path, monkeypatch):
    """'remote' SL source: a replay-actor fleet pushes decoded step lists
    through the Adapter (token 'replay'); dataloader workers pull and fill
    the shared batch (reference sl_dataloader.py remote mode +
    replay_actor.py push)."""
    monkeypatch.chdir(tmp_path)
    from distar_amd.data.adapter import Adapter
    from distar_amd.data.coordinator import Coordinator
    from distar_amd.data.sl_dataloader import SLDataloader
    from distar_amd.lib.consts import fake_step_data
    torch.manual_seed(0)
    coord = Coordinator().run()
    loader = None
    try:
        producer = Adapter(coordinator_port=coord.port)
        for _ in range(3):
            steps = [fake_step_data(train=True, entity_num=32, randomize=True)
                     for _ in range(4)]
            producer.push(steps, token='replay', fs_type='nppickle')
        cfg = Config({
            'learner': {'use_cuda': False, 'use_distributed': False,
                        'data': {'source': 'remote', 'train_data_file': 'none',
                                 'batch_size': 2, 'trajectory_length': 3,
                                 'num_workers': 2, 'epochs': 1}},
            'communication': {'coordinator_ip': '127.0.0.1',
                              'coordinator_port': coord.port}})
        loader = SLDataloader(cfg)
        batch = next(loader)
        assert batch['traj_lens'] == [3, 3]
        assert all(batch['new_episodes'])
        assert batch['entity_num'].shape[0] == 6
    finally:
        if loader is not None:
            loader.close()
        coord.close()


def test_config_backup_written(tmp_path, monkeypatch):
    """Learners back up the fully-merged config into the experiment dir
    (reference bin/rl_train.py:27-42)."""
    monkeypatch.chdir(tmp_path)
    import yaml
    from distar_amd.learner.sl_learner import SLLearner
    cfg = Config({'learner': {'job_type': 'fake', 'use_cuda': False,
                              'data': {'batch_size': 2, 'trajectory_length': 3}},
                  'common': {'experiment_name': 'test_cfg_backup',
                             'type': 'train'}})
    learner = SLLearner(cfg)
    path = os.path.join(learner._exp_dir, 'config_backup',
                        'SLLearner_whole_config.yaml')
    assert os.path.isfile(path)
    loaded = yaml.safe_load(open(path))
    assert loaded['learner']['data']['batch_size'] == 2


@pytest.mark.timeout(600)
def test_rl_learner_param_norm_streams(tmp_path, monkeypatch):
    """save_grad logs per-parameter grad/weight/clipped norms every
    save_log_freq iterations (reference rl_learner.py:118-130)."""
    monkeypatch.chdir(tmp_path)
    import json
    from distar_amd.learner.rl_learner import RLLearner
    torch.manual_seed(0)
    cfg = Config({'learner': {'player_id': 'MP0', 'job_type': 'fake',
                              'use_cuda': False, 'use_amp': False,
                              'save_grad': True, 'save_log_freq': 1,
                              'data': {'batch_size': 2, 'trajectory_length': 3},
                              'hook': {'after_iter': {
                                  'log_show': {'ext_args': {'freq': 1000}}}}},
                  'common': {'experiment_name': 'test_grad_stream',
                             'type': 'train'},
                  'model': {'enable_baselines': ['winloss']}})
    learner = RLLearner(cfg)
    learner.run(max_iterations=1)
    learner._scalar_logger.flush()
    files = glob.glob(os.path.join(learner._exp_dir, 'log', '*.jsonl'))
    assert files
    keys = {json.loads(l)['key'] for f in files for l in open(f)}
    assert any(k.startswith('grad/') for k in keys)
    assert any(k.startswith('clip_grad/') for k in keys)
    assert any(k.startswith('param/') for k in keys)


def test_sl_hidden_state_lane_reset():
    """reset_hidden_state zeroes exactly the new-episode lanes and detaches
    the rest (reference sl_learner.py:31-36)."""
    from distar_amd.learner.sl_learner import SLLearner
    torch.manual_seed(0)
    cfg = Config({'learner': {'job_type': 'fake', 'use_cuda': False,
                              'data': {'batch_size': 3, 'trajectory_length': 2}},
                  'common': {'experiment_name': 'test_lane_reset',
                             'type': 'train'}})
    learner = SLLearner(cfg)
    for l in range(learner.num_layers):
        h = torch.randn(3, learner.hidden_size, requires_grad=True) * 1
        c = torch.randn(3, learner.hidden_size, requires_grad=True) * 1
        learner.hidden_state[l] = (h + 0, c + 0)     # non-leaf, grad-tracked
    before = [tuple(t.clone() for t in hc) for hc in learner.hidden_state]
    learner.reset_hidden_state(torch.tensor([True, False, True]))
    for l in range(learner.num_layers):
        h, c = learner.hidden_state[l]
        assert not h.requires_grad and not c.requires_grad    # detached
        assert (h[0] == 0).all() and (h[2] == 0).all()
        assert (c[0] == 0).all() and (c[2] == 0).all()
        torch.testing.assert_close(h[1], before[l][0][1])     # lane 1 kept
        torch.testing.assert_close(c[1], before[l][1][1])


@pytest.mark.timeout(600)
def test_rl_value_pretrain_phase_switch(tmp_path, monkeypatch):
    """value_pretrain_iters: critic-only updates for N iterations, then the
    policy unfreezes (reference rl_learner.py:147-172)."""
    monkeypatch.chdir(tmp_path)
    from distar_amd.learner.rl_learner import RLLearner
    torch.manual_seed(0)
    cfg = Config({'learner': {'player_id': 'MP0', 'job_type': 'fake',
                              'use_cuda': False, 'use_amp': False,
                              'value_pretrain_iters': 1,
                              'data': {'batch_size': 2, 'trajectory_length': 3},
                              'hook': {'after_iter': {
                                  'log_show': {'ext_args': {'freq': 1000}}}}},
                  'common': {'experiment_name': 'test_vp_switch',
                             'type': 'train'},
                  'model': {'enable_baselines': ['winloss']}})
    learner = RLLearner(cfg)
    m = getattr(learner._model, 'module', learner._model)
    pol_before = m.policy.action_type_head.action_fc.layer2[0].weight.detach().clone()
    learner.run(max_iterations=1)
    # iteration 1 ran under pretrain: policy untouched, loss was critic-only
    torch.testing.assert_close(
        m.policy.action_type_head.action_fc.layer2[0].weight, pol_before)
    assert learner._remain_value_pretrain_iters == 0
    learner.run(max_iterations=2)
    assert learner._remain_value_pretrain_iters == -1
    assert not learner._loss.only_update_value
    assert not m.only_update_baseline
    # policy now moves
    assert not torch.equal(
        m.policy.action_type_head.action_fc.layer2[0].weight, pol_before)
    learner._dataloader.close()
