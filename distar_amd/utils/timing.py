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
