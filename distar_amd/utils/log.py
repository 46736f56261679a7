"""Logging/observability: text logger, scalar sink, windowed variable records.

Functional parity with the reference's `ctools/utils/log_helper.py:23-751`
(build_logger -> (TextLogger, scalar logger, VariableRecord), meters,
pretty_print, the grouped AlphaStar variable table).  The reference emits
TensorBoard event files via tensorboardX; this image has no tensorboard
package, so the scalar sink writes newline-delimited JSON
(`{step, key, value}` per line) with the same `register_var/add_scalar` API —
trivially convertible to TB offline.
"""
import json
import logging
import os
import time
from collections import defaultdict, deque


def pretty_print(result, direct_print=True):
    cleaned = {k: v for k, v in result.items() if v is not None}
    string = json.dumps(cleaned, indent=2, default=str)
    if direct_print:
        print(string)
    return string


class TextLogger:
    def __init__(self, path, name='default'):
        self.logger = logging.getLogger(name)
        self.logger.setLevel(logging.INFO)
        if not self.logger.handlers:
            os.makedirs(path, exist_ok=True)
            fh = logging.FileHandler(os.path.join(path, f'{name}.txt'))
            fh.setFormatter(logging.Formatter('[%(asctime)s][%(levelname)s] %(message)s'))
            self.logger.addHandler(fh)
            sh = logging.StreamHandler()
            sh.setFormatter(logging.Formatter('[%(asctime)s] %(message)s'))
            self.logger.addHandler(sh)

    def info(self, msg):
        self.logger.info(msg)

    error = info


class ScalarLogger:
    """Scalar sink with a tensorboard-like API: JSONL (greppable) + a real
    TensorBoard event file (utils/tb.py — drop-in for the reference's
    tensorboardX streams, `ctools/utils/log_helper.py:23-64`)."""

    def __init__(self, path, name='scalars', tensorboard=True):
        os.makedirs(path, exist_ok=True)
        self._file = open(os.path.join(path, f'{name}.jsonl'), 'a')
        self._vars = set()
        self._tb = None
        if tensorboard:
            try:
                from .tb import SummaryWriter
                self._tb = SummaryWriter(os.path.join(path, 'tb', name))
            except Exception:  # noqa: BLE001 - TB stream is best-effort
                self._tb = None

    def register_var(self, name):
        self._vars.add(name)

    def add_scalar(self, name, value, global_step=0):
        self._file.write(json.dumps({'step': int(global_step), 'key': name,
                                     'value': float(value), 't': time.time()}) + '\n')
        if self._tb is not None:
            self._tb.add_scalar(name, value, global_step)

    def flush(self):
        self._file.flush()
        if self._tb is not None:
            self._tb.flush()

    def close(self):
        self._file.close()
        if self._tb is not None:
            self._tb.close()


class AverageMeter:
    def __init__(self, length=1000):
        self._data = deque(maxlen=length)

    def update(self, value):
        self._data.append(float(value))

    @property
    def val(self):
        return self._data[-1] if self._data else 0.

    @property
    def avg(self):
        return sum(self._data) / len(self._data) if self._data else 0.


class MoveAverageMeter:
    def __init__(self, decay=0.99):
        self.decay = decay
        self._val = None

    def update(self, value):
        value = float(value)
        self._val = value if self._val is None else \
            self.decay * self._val + (1 - self.decay) * value

    @property
    def val(self):
        return self._val if self._val is not None else 0.

    avg = val


class EmaMeter(MoveAverageMeter):
    pass


class VariableRecord:
    """Windowed running means + text table (reference log_helper.VariableRecord)."""

    def __init__(self, length=1000):
        self.length = length
        self.var_dict = {}

    def register_var(self, name, length=None):
        self.var_dict[name] = AverageMeter(length or self.length)

    def update_var(self, info):
        for k, v in info.items():
            if k not in self.var_dict:
                self.register_var(k)
            try:
                self.var_dict[k].update(float(v))
            except (TypeError, ValueError):
                pass

    def get_var(self, name):
        return self.var_dict[name]

    def get_vars_text(self, keys=None):
        keys = keys or sorted(self.var_dict.keys())
        rows = []
        for k in keys:
            if k in self.var_dict:
                m = self.var_dict[k]
                rows.append(f'  {k:<40s} val {m.val:>12.6f}  avg {m.avg:>12.6f}')
        return '\n'.join(rows)


class AlphaStarVarRecord(VariableRecord):
    """Grouped RL variable tables (reference log_helper.AlphaStarVarRecord):
    renders the {field} x {reward, value, td, pg, upgo, entropy, kl} grid of
    registered variables as one aligned text block."""

    FIELDS = ('winloss', 'build_order', 'built_unit', 'effect', 'upgrade', 'battle')
    COLUMNS = ('reward', 'value', 'td', 'action_type', 'delay', 'queued',
               'selected_units', 'target_unit', 'target_location', 'total')

    def get_star_text(self):
        lines = ['  ' + f'{"field":<14s}' + ' '.join(f'{c:>14s}' for c in self.COLUMNS)]
        for field in self.FIELDS:
            row = [f'  {field:<14s}']
            for col in self.COLUMNS:
                key = f'{field}/{col}'
                row.append(f'{self.var_dict[key].avg:14.5f}' if key in self.var_dict
                           else ' ' * 14)
            lines.append(' '.join(row))
        for group in ('upgo', 'entropy', 'kl'):
            row = [f'  {group:<14s}']
            for col in self.COLUMNS:
                key = f'{group}/{col}'
                row.append(f'{self.var_dict[key].avg:14.5f}' if key in self.var_dict
                           else ' ' * 14)
            lines.append(' '.join(row))
        return '\n'.join(lines)

    def get_vars_text(self, keys=None):
        grouped = {k for k in self.var_dict
                   if '/' in k and k.split('/')[0] in self.FIELDS + ('upgo', 'entropy', 'kl')}
        flat = super().get_vars_text([k for k in sorted(self.var_dict) if k not in grouped])
        return self.get_star_text() + '\n' + flat


def build_logger(cfg, name='default', rank=0):
    """-> (TextLogger, ScalarLogger, VariableRecord); loggers only on rank 0
    (reference log_helper.build_logger)."""
    path = os.path.join(cfg.common.experiment_dir if 'common' in cfg
                        and 'experiment_dir' in cfg.common else 'experiments',
                        cfg.common.experiment_name if 'common' in cfg
                        and 'experiment_name' in cfg.common else 'default', 'log')
    if rank == 0:
        logger = TextLogger(path, name=name)
        scalar_logger = ScalarLogger(path, name=name)
    else:
        logger, scalar_logger = None, None
    var_type = cfg.get('learner', {}).get('var_record_type') if hasattr(cfg, 'get') else None
    record = AlphaStarVarRecord() if var_type == 'alphastar' else VariableRecord()
    return logger, scalar_logger, record
