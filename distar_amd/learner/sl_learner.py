"""Supervised learner (reference `agent/default/sl_learner.py:13-85`):
persistent LSTM hidden state carried across iterations per batch lane,
zeroed on episode boundaries; first 5 optimizer steps skipped; gradient
clip + Adam + warmup LR.

MI355X: forward+loss run under bf16 autocast (MFMA GEMMs); optimizer and
grad clip stay fp32.
"""
import torch

from .base_learner import BaseLearner
from ..data.fake_dataloader import FakeSLDataloader
from ..losses.sl_loss import SupervisedLoss
from ..models.alphastar.model import Model
from ..utils.grad_clip import build_grad_clip


class SLLearner(BaseLearner):
    _name = 'SLLearner'

    def _setup_model(self):
        self._model = Model(self._whole_cfg, temperature=1.0)
        self._grad_clip = build_grad_clip(self._whole_cfg.learner.grad_clip)
        self.num_layers = self._model.cfg.encoder.core_lstm.num_layers
        self.hidden_size = self._model.cfg.encoder.core_lstm.hidden_size
        zero = torch.zeros(self._whole_cfg.learner.data.batch_size, self.hidden_size)
        if self._use_cuda:
            zero = zero.cuda()
        self.hidden_state = [(zero, zero) for _ in range(self.num_layers)]
        self.ignore_step = 0
        # debug mode (reference sl_learner.py:25-29,55-60): EMA-track the five
        # per-head losses; a 10x spike after iteration 200 snapshots the bad
        # batch + checkpoint for offline repro
        self.debug = self._whole_cfg.learner.get('debug', False)
        if self.debug:
            self.debug_loss = {k: 0. for k in (
                'action_type_loss', 'delay_loss', 'selected_units_loss_norm',
                'target_unit_loss', 'target_location_loss')}
            self.debug_min_iter = self._whole_cfg.learner.get('debug_min_iter', 200)

    def reset_hidden_state(self, new_episodes):
        for l in range(self.num_layers):
            h = self.hidden_state[l][0].clone().detach()
            c = self.hidden_state[l][1].clone().detach()
            h[new_episodes] = 0
            c[new_episodes] = 0
            self.hidden_state[l] = (h, c)

    def _setup_loss(self):
        self._loss = SupervisedLoss(self._whole_cfg)

    def _setup_dataloader(self):
        if self._whole_cfg.learner.job_type == 'train':
            from ..data.sl_dataloader import SLDataloader
            self._dataloader = SLDataloader(self._whole_cfg)
        else:
            self._dataloader = FakeSLDataloader(
                self._whole_cfg, device=self._device if self._use_cuda else None)

    def _train(self, data):
        with self._timer:
            new_episodes = data.pop('new_episodes')
            self.reset_hidden_state(new_episodes)
            with torch.autocast('cuda', dtype=torch.bfloat16, enabled=self._use_amp):
                logits, infer_action_info, hidden_state = self._model.sl_train(
                    **data, hidden_state=self.hidden_state)
                log_vars = self._loss.compute_loss(
                    logits, data['action_info'], data['action_mask'],
                    data['selected_units_num'], data['entity_num'], infer_action_info)
            loss = log_vars['total_loss']
        self._log_buffer['forward_time'] = self._timer.value
        if self.debug:
            self._debug_check(data, log_vars, logits)

        with self._timer:
            if self.ignore_step > 5:
                self._optimizer.zero_grad(set_to_none=True)
                loss.backward()
                if self._use_distributed:
                    self._model.sync_gradients()
                gradient = self._grad_clip.apply(self._model.parameters())
                self._optimizer.step()
                self._lr_scheduler.step()
            else:
                gradient = 0.
            self.ignore_step += 1
        self.hidden_state = [(h.detach(), c.detach()) for h, c in hidden_state]
        self._log_buffer['gradient'] = gradient
        self._log_buffer['backward_time'] = self._timer.value
        self._log_buffer.update({k: (v.item() if isinstance(v, torch.Tensor) else v)
                                 for k, v in log_vars.items()})

    def _debug_check(self, data, log_vars, logits):
        """Snapshot spiking batches (reference sl_learner.py:55-60)."""
        for k in self.debug_loss:
            prev = self.debug_loss[k]
            cur = float(log_vars[k])
            self.debug_loss[k] = prev * 0.95 + cur * 0.05
            if prev > 0 and cur > prev * 10 and self.last_iter.val > self.debug_min_iter:
                self.save_checkpoint()
                import os
                path = os.path.join(
                    self._exp_dir,
                    f'debug_{k}_iter_{self.last_iter.val}_rank_{self._rank}.pth')
                torch.save({'data': data, 'hidden_state': self.hidden_state,
                            'log_vars': {n: float(v) for n, v in log_vars.items()},
                            'logits': {n: (v.detach().cpu() if isinstance(v, torch.Tensor) else v)
                                       for n, v in logits.items()}},
                           path)
                self.info(f'[debug] {k} spiked {cur:.3g} (ema {prev:.3g}); '
                          f'snapshot -> {path}')
