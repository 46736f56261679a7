"""Gradient clipping strategies (reference `ctools/torch_utils/grad_clip.py:19-151`):
none / max_norm (EMA-normalized) / momentum_norm (per-param EMA) /
clip_value (second-moment clamp) / clip_const / pytorch_norm.

On GPU the norm computations use torch's fused multi-tensor paths
(`torch.nn.utils.clip_grad_norm_` with foreach=True) — one kernel per dtype
group rather than a per-parameter loop.
"""
import os

import torch


class GradClip:
    def __init__(self, clip_type, threshold=1.0, norm_type=2, begin_step=200,
                 ignore_threshold=3.0):
        assert clip_type in ('max_norm', 'clip_value', 'none', 'clip_const',
                             'pytorch_norm', 'momentum_norm')
        self.clip_type = clip_type
        self.threshold = threshold
        self.norm_type = norm_type
        self.begin_step = begin_step
        self.ignore_threshold = ignore_threshold
        self.step = 0
        self._ema = None
        self._state = {}
        self._chunks = None          # K14 chunk-view cache (same grads)
        self._chunks_of = None

    def apply(self, parameters):
        params = [p for p in parameters if p.grad is not None]
        if not params:
            return 0.
        self.step += 1
        if self.clip_type == 'none':
            with torch.no_grad():
                total = torch.norm(torch.stack(
                    [torch.norm(p.grad.detach(), self.norm_type) for p in params]),
                    self.norm_type)
            return total.item()
        if self.clip_type == 'pytorch_norm':
            grads = [p.grad for p in params]
            if (self.norm_type == 2 and grads[0].is_cuda
                    and all(g.is_cuda and g.dtype == torch.float32
                            and g.is_contiguous() for g in grads)
                    and os.environ.get('DISTAR_AMD_DISABLE_HIP') != '1'):
                # K14 hand-written multi-tensor path: one norm kernel + one
                # clip kernel with a DEVICE-resident norm (the scale never
                # round-trips to the host inside the step).  Big tensors are
                # split into 64k-element chunks — the kernel runs one block
                # per chunk, and whole-tensor chunks left only ~300 blocks
                # on a 256-CU chip (measured 5 ms for a 0.2 ms reduction).
                from ..ops import hip_ext
                ops = hip_ext.require()
                chunks = self._chunks
                if chunks is None or self._chunks_of is not grads[0]:
                    chunks = []
                    for g in grads:
                        flat = g.view(-1)
                        n = flat.numel()
                        for off in range(0, n, 65536):
                            chunks.append(flat.narrow(0, off,
                                                      min(65536, n - off)))
                    self._chunks = chunks
                    self._chunks_of = grads[0]
                norm_sq = ops.multi_norm_sq(chunks)
                ops.multi_clip(chunks, norm_sq, float(self.threshold), 1e-6)
                return float(norm_sq.sqrt())
            total = torch.nn.utils.clip_grad_norm_(params, self.threshold,
                                                   norm_type=self.norm_type,
                                                   foreach=True)
            return total.item()
        if self.clip_type == 'clip_const':
            with torch.no_grad():
                for p in params:
                    p.grad.clamp_(-self.threshold, self.threshold)
                total = torch.norm(torch.stack(
                    [torch.norm(p.grad, self.norm_type) for p in params]),
                    self.norm_type)
            return total.item()
        if self.clip_type == 'max_norm':
            # clip against a bias-corrected EMA of the global grad norm —
            # reference semantics (`grad_clip.py:50-74`): beta1=0.95 EMA
            # started at 0, bias correction 1-beta1^step, warmup without
            # clipping for begin_step steps, EMA always fed the raw norm.
            beta1 = 0.95
            with torch.no_grad():
                total = torch.norm(torch.stack(
                    [torch.norm(p.grad, self.norm_type) for p in params]),
                    self.norm_type).item()
                if self._ema is None:
                    self._ema = 0.0
                if self.step > self.begin_step:
                    bias_correction = 1 - beta1 ** self.step
                    clip_coef = (self._ema / bias_correction) * self.threshold \
                        / (total + 1e-6)
                    if clip_coef < 1:
                        torch._foreach_mul_([p.grad for p in params], clip_coef)
                self._ema = beta1 * self._ema + (1 - beta1) * total
            return total
        if self.clip_type == 'momentum_norm':
            # per-parameter EMA-normalized clip (reference SL default,
            # `grad_clip.py:75-109`).  Reference *intended* semantics: no clip
            # on the first step (EMA seeded with the first observed norms),
            # from step 2 scale each grad so its norm <= threshold * ema, then
            # update ema with the POST-clip norm at 0.99/0.01.  (The reference
            # code has an append-instead-of-assign bug that leaves the EMA
            # None forever, i.e. it never clips — see PARITY.md.)
            # Fully vectorized: one fused norm kernel + tensorized EMA state,
            # ZERO host syncs (the reference loops with .item() per param).
            with torch.no_grad():
                grads = [p.grad for p in params]
                norms = torch.stack(torch._foreach_norm(grads, self.norm_type))
                ema = self._state.get('ema')
                if ema is None or ema.shape != norms.shape:
                    # first step: no clip, seed the EMA with the raw norms
                    self._state['ema'] = norms.clone()
                    post = norms
                else:
                    scale = (self.threshold * ema / (norms + 1e-6)).clamp(max=1.0)
                    torch._foreach_mul_(grads, list(scale.unbind()))
                    post = norms * scale
                    self._state['ema'] = 0.99 * ema + 0.01 * post
                total = torch.norm(post, self.norm_type)
            return total.item()
        if self.clip_type == 'clip_value':
            # Adam-like second-moment clamp
            with torch.no_grad():
                total_sq = 0.
                for i, p in enumerate(params):
                    state = self._state.setdefault(i, torch.zeros_like(p.grad))
                    state.mul_(0.999).addcmul_(p.grad, p.grad, value=0.001)
                    bound = state.sqrt() * self.threshold + 1e-8
                    p.grad.clamp_(-1e9, 1e9)
                    if self.step > self.begin_step:
                        torch.minimum(p.grad, bound, out=p.grad)
                        torch.maximum(p.grad, -bound, out=p.grad)
                    total_sq += torch.norm(p.grad, self.norm_type).item() ** 2
            return total_sq ** 0.5
        raise KeyError(self.clip_type)


def build_grad_clip(cfg):
    return GradClip(cfg.get('type', 'pytorch_norm'),
                    threshold=cfg.get('threshold', 1.0),
                    norm_type=cfg.get('norm_type', 2),
                    begin_step=cfg.get('begin_step', 200),
                    ignore_threshold=cfg.get('ignore_threshold', 3.0))
