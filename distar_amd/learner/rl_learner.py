"""Reinforcement learner (reference `agent/default/rl_learner.py:23-307`):
V-trace learner with value-pretrain phase, staleness stats, model publishing
via LearnerComm hooks, league-ordered resets; Adam(betas=(0,0.99), eps=1e-5).

MI355X: forward+loss under bf16 autocast; gradient sync is the bucketed
overlapped RCCL allreduce; V-trace/UPGO/TD-lambda scans run as HIP kernels.
"""
import torch

from .base_learner import BaseLearner
from .hooks import Hook
from ..data.fake_dataloader import FakeRLDataloader
from ..losses.rl_loss import ReinforcementLoss
from ..models.alphastar.model import Model
from ..parallel.dist import broadcast, get_world_size, is_initialized
from ..utils.grad_clip import build_grad_clip

RL_DEFAULT_BASELINES = ['winloss', 'build_order', 'built_unit', 'battle']


class SendModelHook(Hook):
    def __init__(self, comm, freq=10, **kwargs):
        super().__init__('send_model', **kwargs)
        self.comm = comm
        self.freq = freq

    def __call__(self, engine):
        if engine.rank == 0 and engine.last_iter.val % self.freq == 0:
            self.comm.send_model(engine)


class SendTrainInfoHook(Hook):
    def __init__(self, comm, freq=10, **kwargs):
        super().__init__('send_train_info', **kwargs)
        self.comm = comm
        self.freq = freq

    def __call__(self, engine):
        if engine.rank == 0 and engine.last_iter.val % self.freq == 0:
            self.comm.send_train_info(engine)


class RLLearner(BaseLearner):
    _name = 'RLLearner'

    def _setup_model(self):
        self._model = Model(self._whole_cfg, use_value_network=True)
        self._grad_clip = build_grad_clip(self._whole_cfg.learner.grad_clip)
        self._player_id = self._whole_cfg.learner.get('player_id', 'MP0')
        self._remain_value_pretrain_iters = \
            self._whole_cfg.learner.get('value_pretrain_iters', -1)
        self._reset_value_flag = False
        self._update_config_flag = False
        self._config_update = None
        self._debug_server = None
        # optional per-parameter grad/weight norm streams (reference
        # rl_learner.py:35-47,118-130); JSONL keyed grad/<param> etc.
        self._save_grad = self._whole_cfg.learner.get('save_grad', False)
        self._save_grad_freq = self._whole_cfg.learner.get('save_log_freq', 100)

    def _setup_loss(self):
        self._loss = ReinforcementLoss(self._whole_cfg.learner, self._player_id)

    def _setup_optimizer(self):
        self._optimizer = torch.optim.Adam(
            self.model.parameters(), lr=self._whole_cfg.learner.learning_rate,
            betas=(0.0, 0.99), eps=1e-5, fused=self._use_cuda or None)
        self._lr_scheduler = torch.optim.lr_scheduler.MultiStepLR(
            self._optimizer, milestones=[], gamma=1)

    def _setup_dataloader(self):
        if self._whole_cfg.learner.job_type == 'train':
            from ..data.rl_dataloader import RLDataLoader
            self._dataloader = RLDataLoader(self._whole_cfg)
        else:
            self._dataloader = FakeRLDataloader(
                self._whole_cfg, device=self._device if self._use_cuda else None)

    def _setup_comm_hooks(self, comm):
        freq = self._whole_cfg.communication.get('learner_send_model_freq', 10) \
            if 'communication' in self._whole_cfg else 10
        self.register_hook(SendModelHook(comm, freq=freq, position='after_iter'))
        info_freq = self._whole_cfg.communication.get('learner_send_train_info_freq', 10) \
            if 'communication' in self._whole_cfg else 10
        self.register_hook(SendTrainInfoHook(comm, freq=info_freq, position='after_iter'))

    def step_value_pretrain(self):
        if self._remain_value_pretrain_iters > 0:
            self._loss.only_update_value = True
            self._remain_value_pretrain_iters -= 1
            if isinstance(self._model, torch.nn.Module):
                m = getattr(self._model, 'module', self._model)
                m.only_update_baseline = True
        elif self._remain_value_pretrain_iters == 0:
            self._loss.only_update_value = False
            self._remain_value_pretrain_iters = -1
            m = getattr(self._model, 'module', self._model)
            m.only_update_baseline = False
            self.info('value pretrain done, policy updates enabled')

    # ---------------------------------------------------------------- debug
    def start_debug_server(self, host='127.0.0.1', port=None):
        """Live-control endpoints (reference rl_learner.py:263-287 serves a
        Flask debug app): POST /learner/update_config {overrides} deep-merges
        config and rebuilds the loss at the next iteration boundary;
        /learner/reset_value reinitializes + rebroadcasts the value networks;
        /learner/reset_comm restarts the dataloader's adapter pulls."""
        from ..utils.http import JsonHttpServer

        def _update_config(body):
            self._config_update = body.get('overrides', {})
            self._update_config_flag = True
            return {'done': True}

        def _reset_value(body):
            self._reset_value_flag = True
            return {'done': True}

        def _reset_comm(body):
            loader = getattr(self, '_dataloader', None)
            if loader is not None and hasattr(loader, 'reset_comm'):
                loader.reset_comm()
            return {'done': True}

        self._debug_server = JsonHttpServer(
            {'/learner/update_config': _update_config,
             '/learner/reset_value': _reset_value,
             '/learner/reset_comm': _reset_comm}, host=host, port=port)
        self._debug_server.start()
        self.info(f'debug server on {host}:{self._debug_server.port}')
        return self._debug_server

    def _apply_debug_flags(self):
        if self._update_config_flag:
            self._update_config_flag = False
            from ..utils.config import Config, deep_merge_dicts
            if self._config_update:
                self._whole_cfg = deep_merge_dicts(
                    self._whole_cfg, Config(self._config_update))
            self._setup_loss()
            self.info(f'config updated: {self._config_update}')
            self._config_update = None
        if self._reset_value_flag:
            self._reset_value_flag = False
            self.reset_value()

    def _train(self, data):
        self._apply_debug_flags()
        with self._timer:
            self.step_value_pretrain()
            model_last_iter = data.pop('model_last_iter', None)
            data.pop('aux_type', None)
            if model_last_iter is not None and self._remain_value_pretrain_iters <= 0:
                iter_diff = (self.last_iter.val - model_last_iter).float()
                self._log_buffer['staleness'] = iter_diff.mean().item()
                self._log_buffer['staleness_max'] = iter_diff.max().item()
            with torch.autocast('cuda', dtype=torch.bfloat16, enabled=self._use_amp):
                model_output = self._model.rl_learner_forward(**data)
                if self._whole_cfg.learner.get('use_dapo', False):
                    model_output['successive_logit'] = data['successive_logit']
                log_vars = self._loss.compute_loss(model_output)
            loss = log_vars['total_loss']
        self._log_buffer['forward_time'] = self._timer.value

        with self._timer:
            self._optimizer.zero_grad(set_to_none=True)
            loss.backward()
            if self._use_distributed:
                self._model.sync_gradients()
            log_params = self._save_grad and self._rank == 0 and \
                self.last_iter.val % self._save_grad_freq == 0
            if log_params:
                with torch.no_grad():
                    for k, p in self._model.named_parameters():
                        if p.grad is not None:
                            self._scalar_logger.add_scalar(
                                f'grad/{k}', p.grad.norm().item(),
                                global_step=self.last_iter.val)
                            self._scalar_logger.add_scalar(
                                f'param/{k}', p.data.norm().item(),
                                global_step=self.last_iter.val)
            gradient = self._grad_clip.apply(self._model.parameters())
            if log_params:
                with torch.no_grad():
                    for k, p in self._model.named_parameters():
                        if p.grad is not None:
                            self._scalar_logger.add_scalar(
                                f'clip_grad/{k}', p.grad.norm().item(),
                                global_step=self.last_iter.val)
            self._optimizer.step()
        self._log_buffer['gradient'] = gradient
        self._log_buffer['backward_time'] = self._timer.value
        self._log_buffer.update({k: (v.item() if isinstance(v, torch.Tensor) else v)
                                 for k, v in log_vars.items()})

    # league-ordered value reset (reference rl_learner.py:226-261)
    def reset_value(self):
        m = getattr(self._model, 'module', self._model)
        for net in m.value_networks.values():
            for layer in net.modules():
                if isinstance(layer, torch.nn.Linear):
                    layer.reset_parameters()
        if is_initialized() and get_world_size() > 1:
            for p in m.value_networks.parameters():
                broadcast(p.data, src=0)
        self.info('value networks reset')
