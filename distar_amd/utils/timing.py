"""Timers.

Functional parity with the reference's `ctools/utils/time_helper.py:45-215`
(EasyTimer with cuda-event timing).  On ROCm, `torch.cuda.Event` is a
hipEvent pair — the same mechanism the reference uses on CUDA.
"""
import time

import torch


class EasyTimer:
    """Context-manager timer; `.value` is seconds of the last block."""

    def __init__(self, cuda=True):
        self.cuda = cuda and torch.cuda.is_available()
        self.value = 0.0
        if self.cuda:
            self._start_ev = torch.cuda.Event(enable_timing=True)
            self._end_ev = torch.cuda.Event(enable_timing=True)

    def __enter__(self):
        if self.cuda:
            self._start_ev.record()
        else:
            self._start = time.perf_counter()
        return self

    def __exit__(self, *exc):
        if self.cuda:
            self._end_ev.record()
            self._end_ev.synchronize()
            self.value = self._start_ev.elapsed_time(self._end_ev) / 1000.0
        else:
            self.value = time.perf_counter() - self._start
        return False


class Stopwatch:
    """Hierarchical wall-time profiler (functional parity with pysc2's
    `lib/stopwatch.py`, which the reference wraps around obs-transform and
    protocol calls): nested `with sw("name")` blocks accumulate under
    dot-joined paths; `str(sw)` renders a sorted report.  Disabled by
    default — negligible overhead until `sw.enable()` (or the
    DISTAR_AMD_STOPWATCH=1 env) turns it on."""

    def __init__(self, enabled=None):
        import os
        self._enabled = (os.environ.get('DISTAR_AMD_STOPWATCH') == '1'
                         if enabled is None else enabled)
        self._times = {}        # path -> [count, total_seconds]
        self._stack = []

    def enable(self):
        self._enabled = True

    def disable(self):
        self._enabled = False

    def clear(self):
        self._times.clear()

    class _Block:
        __slots__ = ('sw', 'name', 't0')

        def __init__(self, sw, name):
            self.sw = sw
            self.name = name

        def __enter__(self):
            if self.sw._enabled:
                self.sw._stack.append(self.name)
                self.t0 = time.perf_counter()
            return self

        def __exit__(self, *exc):
            if self.sw._enabled:
                dt = time.perf_counter() - self.t0
                path = '.'.join(self.sw._stack)
                rec = self.sw._times.setdefault(path, [0, 0.0])
                rec[0] += 1
                rec[1] += dt
                self.sw._stack.pop()
            return False

    def __call__(self, name):
        return Stopwatch._Block(self, name)

    def decorate(self, name=None):
        def wrap(fn):
            import functools
            label = name or fn.__name__

            @functools.wraps(fn)
            def inner(*args, **kwargs):
                if not self._enabled:
                    return fn(*args, **kwargs)
                with self(label):
                    return fn(*args, **kwargs)
            return inner
        return wrap

    def __str__(self):
        if not self._times:
            return 'stopwatch: no samples'
        lines = [f'{"path":<40s} {"calls":>8s} {"total s":>10s} {"avg ms":>10s}']
        for path in sorted(self._times):
            n, total = self._times[path]
            lines.append(f'{path:<40s} {n:8d} {total:10.3f} {total/n*1e3:10.2f}')
        return '\n'.join(lines)


sw = Stopwatch()
