"""Loader for the in-tree HIP/CDNA4 extension (`_hip_ops*.so`).

The extension is built IN-TREE by `python setup_hip.py build` (driven by
`__graft_entry__.build()`) with `hipcc --offload-arch=gfx950`, so the .so
travels to GPU boxes with the repo snapshot.

Policy: on a GPU (`tensor.is_cuda`), the HIP path is mandatory — a missing
extension raises ImportError rather than silently falling back to eager
(round-end native-code checks require the .so to actually load).  On CPU the
eager fallbacks run.  Set DISTAR_AMD_DISABLE_HIP=1 to force eager on GPU
(A/B comparisons only).
"""
import importlib
import os
import sys

_EXT = None
_TRIED = False

_HERE = os.path.dirname(os.path.abspath(__file__))


def _load():
    global _EXT, _TRIED
    if _TRIED:
        return _EXT
    _TRIED = True
    if _HERE not in sys.path:
        sys.path.insert(0, _HERE)
    try:
        _EXT = importlib.import_module('_hip_ops')
    except ImportError:
        _EXT = None
    return _EXT


def available():
    return _load() is not None


def maybe_ext(tensor):
    """Return the extension module for CUDA(=ROCm) tensors, None on CPU."""
    if not tensor.is_cuda:
        return None
    if os.environ.get('DISTAR_AMD_DISABLE_HIP') == '1':
        return None
    ext = _load()
    if ext is None:
        raise ImportError(
            'distar_amd HIP extension (_hip_ops) is not built but a GPU tensor '
            'reached a HIP-op call site. Build it in-tree with '
            '`python setup_hip.py build` (hipcc --offload-arch=gfx950); '
            'refusing to fall back to eager on GPU.')
    return ext


def require():
    """The extension module, or ImportError (HIP-op call sites on GPU)."""
    ext = _load()
    if ext is None:
        raise ImportError(
            'distar_amd HIP extension (_hip_ops) is not built; run '
            '`python setup_hip.py build` (hipcc --offload-arch=gfx950).')
    return ext
