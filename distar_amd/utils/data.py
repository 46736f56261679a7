"""Tensor-tree helpers: device moves, dtype casts, collate functions.

Functional parity with the reference's `ctools/torch_utils/data_helper.py`
(to_device/to_dtype/to_tensor) and `ctools/data/collate_fn.py`
(default_collate / default_collate_with_dim — the dim-aware variant skips
keys missing from individual dicts, which is how the RL collate stacks T+1
observation frames against T action frames).
"""
import numbers
from collections.abc import Mapping, Sequence

import numpy as np
import torch


def to_device(item, device, non_blocking=False):
    if isinstance(item, torch.Tensor):
        return item.to(device, non_blocking=non_blocking)
    if isinstance(item, Mapping):
        return {k: to_device(v, device, non_blocking) for k, v in item.items()}
    if isinstance(item, tuple):
        return tuple(to_device(v, device, non_blocking) for v in item)
    if isinstance(item, list):
        return [to_device(v, device, non_blocking) for v in item]
    return item


def to_dtype(item, dtype):
    if isinstance(item, torch.Tensor):
        return item.to(dtype)
    if isinstance(item, Mapping):
        return {k: to_dtype(v, dtype) for k, v in item.items()}
    if isinstance(item, (list, tuple)):
        return type(item)(to_dtype(v, dtype) for v in item)
    return item


def to_tensor(item, dtype=None):
    if isinstance(item, torch.Tensor):
        return item if dtype is None else item.to(dtype)
    if isinstance(item, np.ndarray):
        t = torch.from_numpy(item)
        return t if dtype is None else t.to(dtype)
    if isinstance(item, Mapping):
        return {k: to_tensor(v, dtype) for k, v in item.items()}
    if isinstance(item, (list, tuple)):
        return type(item)(to_tensor(v, dtype) for v in item)
    if isinstance(item, numbers.Number):
        return torch.tensor(item, dtype=dtype)
    return item


def to_share_memory(item):
    if isinstance(item, torch.Tensor):
        return item.share_memory_()
    if isinstance(item, Mapping):
        return {k: to_share_memory(v) for k, v in item.items()}
    if isinstance(item, (list, tuple)):
        return type(item)(to_share_memory(v) for v in item)
    return item


def default_collate(batch):
    """Stack a list of samples along a new leading batch dim."""
    elem = batch[0]
    if isinstance(elem, torch.Tensor):
        if elem.shape == (1,):
            return torch.cat(batch, 0)
        return torch.stack(batch, 0)
    if isinstance(elem, np.ndarray):
        return default_collate([torch.as_tensor(b) for b in batch])
    if isinstance(elem, float):
        return torch.tensor(batch, dtype=torch.float32)
    if isinstance(elem, bool):
        return torch.tensor(batch, dtype=torch.bool)
    if isinstance(elem, int):
        return torch.tensor(batch, dtype=torch.int64)
    if isinstance(elem, str):
        return batch
    if isinstance(elem, Mapping):
        return {k: default_collate([d[k] for d in batch]) for k in elem}
    if isinstance(elem, tuple) and hasattr(elem, '_fields'):
        return type(elem)(*(default_collate(s) for s in zip(*batch)))
    if isinstance(elem, Sequence):
        return [default_collate(s) for s in zip(*batch)]
    raise TypeError(type(elem))


def default_collate_with_dim(batch, device='cpu', dim=0, k=None):
    """Stack along ``dim``; mapping keys missing from some dicts are collated
    over the dicts that have them (T+1 obs vs T actions in RL trajectories)."""
    elem = batch[0]
    if isinstance(elem, torch.Tensor):
        return torch.stack(batch, dim=dim).to(device=device)
    if isinstance(elem, np.ndarray):
        return default_collate_with_dim([torch.as_tensor(b, device=device) for b in batch],
                                        device=device, dim=dim)
    if isinstance(elem, (float, int)):
        return torch.tensor(batch, device=device)
    if isinstance(elem, str):
        return batch
    if isinstance(elem, Mapping):
        return {key: default_collate_with_dim([d[key] for d in batch if key in d],
                                              device=device, dim=dim, k=key)
                for key in elem}
    if isinstance(elem, tuple) and hasattr(elem, '_fields'):
        return type(elem)(*(default_collate_with_dim(s, device=device, dim=dim)
                            for s in zip(*batch)))
    if isinstance(elem, Sequence):
        it = iter(batch)
        n = len(next(it))
        if not all(len(e) == n for e in it):
            raise RuntimeError('unequal sequence lengths in batch')
        return [default_collate_with_dim(s, device=device, dim=dim) for s in zip(*batch)]
    raise TypeError(type(elem))


def flat(data):
    """Flatten the leading (T, B) dims of every tensor in a tree."""
    if isinstance(data, torch.Tensor):
        return torch.flatten(data, start_dim=0, end_dim=1)
    if isinstance(data, Mapping):
        return {k: flat(v) for k, v in data.items()}
    if isinstance(data, Sequence) and not isinstance(data, str):
        return [flat(v) for v in data]
    raise TypeError(type(data))


class CudaFetcher:
    """Background H2D prefetcher (reference `ctools/torch_utils/data_helper.py:203-231`):
    a producer thread pulls from ``data_source``, stages each batch to
    ``device`` on its own HIP stream (pinned-host copies overlap compute on
    the default stream), and parks results in a bounded queue.

    The RL learner path uses the richer ``data.rl_dataloader._DeviceStager``
    (per-batch events); this class is the drop-in generic fetcher for any
    iterator."""

    def __init__(self, data_source, device, queue_size=4, sleep=0.1):
        import queue as _queue
        import threading
        self._source = data_source
        self._queue = _queue.Queue(maxsize=queue_size)
        self._device = device
        self._sleep = sleep
        self._end_flag = True
        self._stream = None
        self._thread = threading.Thread(target=self._producer, daemon=True)

    def __iter__(self):
        return self

    def __next__(self):
        return self._queue.get()

    def run(self):
        self._end_flag = False
        self._thread.start()
        return self

    def close(self):
        self._end_flag = True

    def _producer(self):
        import time as _time
        use_cuda = str(self._device).startswith('cuda') and torch.cuda.is_available()
        if use_cuda and self._stream is None:
            self._stream = torch.cuda.Stream()
        ctx = torch.cuda.stream(self._stream) if use_cuda else _NullCtx()
        with ctx:
            while not self._end_flag:
                if self._queue.full():
                    _time.sleep(self._sleep)
                    continue
                try:
                    data = next(self._source)
                except StopIteration:
                    self._end_flag = True
                    break
                data = to_device(data, self._device,
                                 non_blocking=use_cuda)
                if use_cuda:
                    self._stream.synchronize()
                self._queue.put(data)


class _NullCtx:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False
