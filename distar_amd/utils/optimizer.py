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
