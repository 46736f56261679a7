"""Extended Adam (reference `ctools/torch_utils/optimizer_util.py:44-318`):
stock Adam plus optional fused per-step gradient clipping/ignoring variants.

On ROCm the underlying step uses torch's multi-tensor (foreach) Adam, so the
whole parameter set updates in a handful of kernels rather than a per-param
loop.
"""
import torch

from .grad_clip import GradClip


class Adam(torch.optim.Adam):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0, optim_type='adam',
                 grad_clip_type=None, clip_value=None, clip_norm_type=2,
                 clip_momentum_timestep=100, ignore_momentum_timestep=100,
                 foreach=True):
        super().__init__(params, lr=lr, betas=betas, eps=eps,
                         weight_decay=weight_decay, foreach=foreach)
        self._clip = None
        if grad_clip_type is not None:
            self._clip = GradClip(grad_clip_type, threshold=clip_value or 1.0,
                                  norm_type=clip_norm_type,
                                  begin_step=clip_momentum_timestep)

    @torch.no_grad()
    def step(self, closure=None):
        if self._clip is not None:
            params = [p for group in self.param_groups for p in group['params']]
            self._clip.apply(params)
        return super().step(closure)


class GradualWarmupScheduler(torch.optim.lr_scheduler.LRScheduler):
    """Linear warm-up to base_lr * multiplier over ``total_epoch`` steps, then
    hand off to ``after_scheduler`` (functional parity with the reference's
    `ctools/torch_utils/lr_scheduler_util.py`; the learners use torch's
    SequentialLR(LinearLR, ...) composition by default, this class is the
    standalone utility)."""

    def __init__(self, optimizer, multiplier, total_epoch, after_scheduler=None):
        if multiplier < 1.0:
            raise ValueError('multiplier must be >= 1.0')
        self.multiplier = multiplier
        self.total_epoch = total_epoch
        self.after_scheduler = after_scheduler
        self.finished = False
        super().__init__(optimizer)

    def get_lr(self):
        if self.last_epoch > self.total_epoch:
            if self.after_scheduler is not None:
                if not self.finished:
                    self.after_scheduler.base_lrs = [
                        b * self.multiplier for b in self.base_lrs]
                    self.finished = True
                return self.after_scheduler.get_last_lr()
            return [b * self.multiplier for b in self.base_lrs]
        if self.multiplier == 1.0:
            return [b * float(self.last_epoch) / self.total_epoch
                    for b in self.base_lrs]
        return [b * ((self.multiplier - 1.) * self.last_epoch / self.total_epoch + 1.)
                for b in self.base_lrs]

    def step(self, epoch=None):
        if self.finished and self.after_scheduler is not None:
            self.after_scheduler.step(
                None if epoch is None else epoch - self.total_epoch)
            self._last_lr = self.after_scheduler.get_last_lr()
        else:
            super().step(epoch)
