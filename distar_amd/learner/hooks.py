"""Priority-ordered learner hook system.

Functional parity with the reference's `ctools/worker/learner/learner_hook.py`
(4 positions: before_run / before_iter / after_iter / after_run; built-ins:
LrSchedulerHook, LoadCkptHook, SaveCkptHook, LogShowHook, LogReduceHook).
LogReduceHook coalesces the whole scalar log buffer into ONE flat allreduce
(the reference reduces each entry separately, `learner_hook.py:271-325` —
dozens of tiny collectives per iteration).
"""
import numbers
import os

import torch

from ..parallel.dist import allreduce, get_world_size, is_initialized

HOOK_POSITIONS = ('before_run', 'before_iter', 'after_iter', 'after_run')


class Hook:
    def __init__(self, name, priority=100, position='after_iter', ext_args=None):
        self.name = name
        self.priority = priority
        self.position = position
        self.ext_args = ext_args or {}

    def __call__(self, engine):
        raise NotImplementedError


class LrSchedulerHook(Hook):
    def __init__(self, **kwargs):
        super().__init__('lr_scheduler', **kwargs)
        self.niter = self.ext_args.get('freq', 1)

    def __call__(self, engine):
        if engine.last_iter.val % self.niter == 0:
            engine.lr_scheduler.step()
        engine.log_buffer['cur_lr'] = engine.lr_scheduler.get_last_lr()[0]


class LoadCkptHook(Hook):
    def __init__(self, **kwargs):
        super().__init__('load_ckpt', position='before_run', **kwargs)

    def __call__(self, engine):
        path = engine.load_path
        if not path or not os.path.exists(path):
            return
        engine.checkpoint_helper.load(
            path, engine.model, optimizer=engine.optimizer,
            last_iter=engine.last_iter,
            state_dict_mask=engine.whole_cfg.learner.get('state_dict_mask', []),
            strict=engine.whole_cfg.learner.get('load_strict', True),
            logger_prints=engine.info)
        # fast-forward the LR scheduler to the restored iteration.  torch
        # warns "scheduler.step() before optimizer.step()" here because no
        # optimizer step has run yet this process — that is exactly the
        # resume situation, so the warning is noise; the schedule lands on
        # last_epoch == last_iter, same as if training had never stopped.
        import warnings
        with warnings.catch_warnings():
            warnings.filterwarnings(
                'ignore', message='.*lr_scheduler.step().*optimizer.step().*')
            for _ in range(engine.last_iter.val):
                engine.lr_scheduler.step()
        engine.info(f'loaded checkpoint {path} (iter {engine.last_iter.val})')


class SaveCkptHook(Hook):
    def __init__(self, **kwargs):
        super().__init__('save_ckpt', **kwargs)
        self.freq = self.ext_args.get('freq', 1000)

    def __call__(self, engine):
        if engine.rank != 0:
            return
        if engine.last_iter.val % self.freq != 0:
            return
        engine.save_checkpoint()


class LogShowHook(Hook):
    def __init__(self, **kwargs):
        kwargs.setdefault('priority', 30)
        super().__init__('log_show', **kwargs)
        self.freq = self.ext_args.get('freq', 100)

    def __call__(self, engine):
        if engine.rank != 0:
            return
        engine.record.update_var(
            {k: v for k, v in engine.log_buffer.items() if isinstance(v, numbers.Number)
             or (isinstance(v, torch.Tensor) and v.numel() == 1)})
        iteration = engine.last_iter.val
        if iteration % self.freq == 0:
            engine.info(f'=== iter {iteration} ===\n' + engine.record.get_vars_text())
            if engine.scalar_logger is not None:
                for k, v in engine.log_buffer.items():
                    if isinstance(v, torch.Tensor) and v.numel() == 1:
                        v = v.item()
                    if isinstance(v, numbers.Number):
                        engine.scalar_logger.add_scalar(k, v, global_step=iteration)
                engine.scalar_logger.flush()
        engine.log_buffer.clear()


class LogReduceHook(Hook):
    """Allreduce the scalar log buffer across ranks as ONE flat tensor."""

    def __init__(self, **kwargs):
        kwargs.setdefault('priority', 10)
        super().__init__('log_reduce', **kwargs)

    def __call__(self, engine):
        if not is_initialized() or get_world_size() == 1:
            return
        keys = sorted(k for k, v in engine.log_buffer.items()
                      if isinstance(v, numbers.Number)
                      or (isinstance(v, torch.Tensor) and v.numel() == 1))
        if not keys:
            return
        device = 'cuda' if torch.cuda.is_available() else 'cpu'
        vals = torch.tensor(
            [float(engine.log_buffer[k].item() if isinstance(engine.log_buffer[k], torch.Tensor)
                   else engine.log_buffer[k]) for k in keys],
            dtype=torch.float32, device=device)
        allreduce(vals, average=True)
        for k, v in zip(keys, vals.tolist()):
            engine.log_buffer[k] = v


def build_learner_hook_by_cfg(cfg):
    """cfg: {position: {hook_name: {'ext_args': {...}, 'priority': int}}}"""
    registry = {
        'lr_scheduler': LrSchedulerHook, 'load_ckpt': LoadCkptHook,
        'save_ckpt': SaveCkptHook, 'log_show': LogShowHook,
        'log_reduce': LogReduceHook,
    }
    hooks = {pos: [] for pos in HOOK_POSITIONS}
    for position, entries in (cfg or {}).items():
        for name, args in (entries or {}).items():
            cls = registry[name]
            kw = dict(args or {})
            kw.setdefault('ext_args', {})
            hook = cls(priority=kw.get('priority', 100), ext_args=kw['ext_args'])
            hook.position = position
            hooks[position].append(hook)
    for pos in hooks:
        hooks[pos].sort(key=lambda h: h.priority)
    return hooks


def add_learner_hook(hooks, hook):
    hooks[hook.position].append(hook)
    hooks[hook.position].sort(key=lambda h: h.priority)
