"""Checkpoint save/load with the reference's exact dict layout.

Functional parity with `ctools/torch_utils/checkpoint_helper.py:34-369`:
`{'model': state_dict, 'optimizer': ..., 'last_iter': int, ...}` files,
prefix add/remove, `state_dict_mask`, `CountVar`, and the `auto_checkpoint`
decorator (save on any exception / SIGINT / SIGTERM around the train loop).
Reference checkpoints load drop-in (§5.4 of SURVEY.md).
"""
import os
import signal
import traceback
from collections import OrderedDict

import torch


class CountVar:
    def __init__(self, init_val=0):
        self._val = init_val

    @property
    def val(self):
        return self._val

    def update(self, val):
        self._val = val

    def add(self, add_num):
        self._val += add_num


def build_checkpoint_helper(cfg=None, rank=0):
    return CheckpointHelper(rank)


class CheckpointHelper:
    def __init__(self, rank=0):
        self._rank = rank

    @staticmethod
    def _remove_prefix(state_dict, prefix='module.'):
        return OrderedDict((k[len(prefix):] if k.startswith(prefix) else k, v)
                           for k, v in state_dict.items())

    @staticmethod
    def _add_prefix(state_dict, prefix='module.'):
        return OrderedDict((prefix + k, v) for k, v in state_dict.items())

    def save(self, path, model, optimizer=None, last_iter=None, last_epoch=None,
             dataset=None, collector_info=None, prefix_op=None, prefix=None, **extra):
        checkpoint = {}
        state_dict = model.state_dict()
        if prefix_op is not None:
            state_dict = {'remove': self._remove_prefix,
                          'add': self._add_prefix}[prefix_op](state_dict, prefix)
        checkpoint['model'] = state_dict
        if optimizer is not None:
            checkpoint['optimizer'] = optimizer.state_dict()
        if last_iter is not None:
            checkpoint['last_iter'] = last_iter.val if isinstance(last_iter, CountVar) else last_iter
        if last_epoch is not None:
            checkpoint['last_epoch'] = last_epoch.val if isinstance(last_epoch, CountVar) else last_epoch
        checkpoint.update(extra)
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        torch.save(checkpoint, path)

    def load(self, path, model, optimizer=None, last_iter=None, last_epoch=None,
             prefix_op=None, prefix=None, strict=True, state_dict_mask=None,
             logger_prints=print, need_torch_load=True, map_location='cpu'):
        checkpoint = torch.load(path, map_location=map_location,
                                weights_only=False) if need_torch_load else path
        state_dict = checkpoint['model'] if 'model' in checkpoint else checkpoint
        if prefix_op is not None:
            state_dict = {'remove': self._remove_prefix,
                          'add': self._add_prefix}[prefix_op](state_dict, prefix)
        if state_dict_mask:
            masked = OrderedDict()
            for k, v in state_dict.items():
                if not any(k.startswith(m) for m in state_dict_mask):
                    masked[k] = v
            state_dict = masked
            strict = False
        missing, unexpected = model.load_state_dict(state_dict, strict=strict)
        if missing:
            logger_prints(f'checkpoint load: missing keys {list(missing)[:8]}... '
                          f'({len(missing)} total)')
        if unexpected:
            logger_prints(f'checkpoint load: unexpected keys {list(unexpected)[:8]}... '
                          f'({len(unexpected)} total)')
        if optimizer is not None and 'optimizer' in checkpoint:
            optimizer.load_state_dict(checkpoint['optimizer'])
        if last_iter is not None and 'last_iter' in checkpoint:
            last_iter.update(checkpoint['last_iter'])
        if last_epoch is not None and 'last_epoch' in checkpoint:
            last_epoch.update(checkpoint['last_epoch'])
        return checkpoint


def auto_checkpoint(func):
    """Save a checkpoint on any exception or termination signal around the
    wrapped method (reference `checkpoint_helper.py:325-369`).  The instance
    must expose `save_checkpoint()`."""
    handled = [getattr(signal, n) for n in
               ('SIGINT', 'SIGTERM', 'SIGHUP', 'SIGQUIT') if hasattr(signal, n)]

    def wrapper(self, *args, **kwargs):
        def handler(signum, frame):
            raise SystemExit(f'signal {signum}')
        prev = {}
        try:
            for sig in handled:
                try:
                    prev[sig] = signal.signal(sig, handler)
                except (ValueError, OSError):
                    pass  # not in main thread
            return func(self, *args, **kwargs)
        except (BaseException,):
            traceback.print_exc()
            try:
                self.save_checkpoint()
                print('[auto_checkpoint] emergency checkpoint saved')
            except Exception:
                traceback.print_exc()
            raise
        finally:
            for sig, h in prev.items():
                try:
                    signal.signal(sig, h)
                except (ValueError, OSError):
                    pass
    return wrapper
