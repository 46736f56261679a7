"""Small structural helpers (functional parity with the reference's
`ctools/utils/default_helper.py` and `ctools/torch_utils/detach.py`,
re-written fresh)."""
from typing import Any, Callable, List, Mapping, Sequence

import torch


def lists_to_dicts(data):
    """[{k: v1}, {k: v2}] -> {k: [v1, v2]} (reference default_helper.py:9)."""
    if not data:
        raise ValueError('empty data')
    if isinstance(data[0], dict):
        return {k: [d[k] for d in data] for k in data[0]}
    if hasattr(data[0], '_fields'):        # namedtuple
        cls = type(data[0])
        return cls(*(list(v) for v in zip(*data)))
    raise TypeError(type(data[0]))


def dicts_to_lists(data: Mapping) -> List[Mapping]:
    """{k: [v1, v2]} -> [{k: v1}, {k: v2}] (reference default_helper.py:50)."""
    if not data:
        raise ValueError('empty data')
    keys = list(data.keys())
    n = len(data[keys[0]])
    return [{k: data[k][i] for k in keys} for i in range(n)]


def override(cls: type) -> Callable:
    """Decorator asserting the decorated method overrides one in ``cls``."""
    def check(method):
        assert method.__name__ in dir(cls), \
            f'{method.__name__} does not override anything in {cls.__name__}'
        return method
    return check


def squeeze(data):
    """Unwrap singleton tuples/lists/dicts (reference default_helper.py:91)."""
    if isinstance(data, (tuple, list)) and len(data) == 1:
        return data[0]
    if isinstance(data, dict) and len(data) == 1:
        return next(iter(data.values()))
    return data


def default_get(data, name, default_value=None, default_fn=None, judge_fn=None):
    """dict get with a lazily-built default and optional validation."""
    if name in data:
        value = data[name]
        if judge_fn is not None and not judge_fn(value):
            raise ValueError(f'invalid value for {name}: {value!r}')
        return value
    return default_fn() if default_fn is not None else default_value


def list_split(data: list, step: int) -> List[list]:
    """Chunk a list into step-sized pieces (last piece may be short)."""
    return [data[i:i + step] for i in range(0, len(data), step)]


def error_wrapper(fn, default_ret, warning_msg='[WARNING] call error'):
    """Call ``fn``; on any exception print ``warning_msg`` and return
    ``default_ret`` (reference default_helper.py:149, used around optional
    backends)."""
    def wrapper(*args, **kwargs):
        try:
            return fn(*args, **kwargs)
        except Exception:  # noqa: BLE001
            print(warning_msg)
            return default_ret
    return wrapper


def get_tensor_data(data: Any) -> Any:
    """Deep-detach a tensor tree into fresh leaves outside any autograd graph
    (reference detach.py get_tensor_data)."""
    if isinstance(data, torch.Tensor):
        return data.detach().clone()
    if data is None:
        return None
    if isinstance(data, dict):
        return {k: get_tensor_data(v) for k, v in data.items()}
    if isinstance(data, Sequence) and not isinstance(data, str):
        return type(data)(get_tensor_data(v) for v in data)
    return data


def seed_everything(seed, deterministic=False):
    """Seed python/numpy/torch (+ cuda) in one call; with ``deterministic``
    also flips torch into deterministic-algorithms mode (the
    reference has no deterministic test mode; this is the rebuild's).

    Returns the seed so callers can log it.
    """
    import os
    import random as _random

    import numpy as _np
    _random.seed(seed)
    _np.random.seed(seed % (2 ** 32))
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
    if deterministic:
        os.environ.setdefault('CUBLAS_WORKSPACE_CONFIG', ':4096:8')
        torch.use_deterministic_algorithms(True, warn_only=True)
        torch.backends.cudnn.deterministic = True     # MIOpen on ROCm
        torch.backends.cudnn.benchmark = False
    return seed
