"""Payload (de)serialization for the trajectory/model transport.

Functional parity with the reference's `ctools/utils/file_helper.py:255-302`
(`dumps`/`loads` with fs_type in {pickle, cPickle, nppickle, torch, pyarrow}
+ lz4 frame compression).  This image has no lz4 module; compression is
zlib-1 (fast) behind the same API, with a magic header so either side can
detect the codec.  'nppickle' converts torch tensors to numpy first —
cheaper to pickle and what the RL trajectory path uses.
"""
import io
import os
import pickle
import sys
import zlib

import numpy as np
import torch

_MAGIC_ZLIB = b'DAZ1'
_MAGIC_RAW = b'DAR0'

# native GIL-released zlib codec (ops/cpp/native_codec.cpp); the pure-python
# zlib module is the fallback when the .so has not been built
_OPS_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), 'ops')
if _OPS_DIR not in sys.path:
    sys.path.insert(0, _OPS_DIR)
try:
    import _native_codec

    def _compress(payload, level):
        return _native_codec.compress(payload, level)

    def _decompress(payload):
        return _native_codec.decompress(payload)
except ImportError:
    def _compress(payload, level):
        return zlib.compress(payload, level)

    def _decompress(payload):
        return zlib.decompress(payload)


def _tensor_to_np(data):
    if isinstance(data, torch.Tensor):
        return {'__np__': data.numpy()} if data.dtype != torch.bfloat16 \
            else {'__npbf16__': data.view(torch.int16).numpy()}
    if isinstance(data, dict):
        return {k: _tensor_to_np(v) for k, v in data.items()}
    if isinstance(data, (list, tuple)):
        return type(data)(_tensor_to_np(v) for v in data)
    return data


def _np_to_tensor(data):
    if isinstance(data, dict):
        if '__np__' in data and len(data) == 1:
            return torch.from_numpy(data['__np__'])
        if '__npbf16__' in data and len(data) == 1:
            return torch.from_numpy(data['__npbf16__']).view(torch.bfloat16)
        return {k: _np_to_tensor(v) for k, v in data.items()}
    if isinstance(data, (list, tuple)):
        return type(data)(_np_to_tensor(v) for v in data)
    return data


def dumps(data, fs_type='nppickle', compress=True, level=1):
    if fs_type in ('pickle', 'cPickle'):
        payload = pickle.dumps(data, protocol=pickle.HIGHEST_PROTOCOL)
    elif fs_type == 'nppickle':
        payload = pickle.dumps(_tensor_to_np(data), protocol=pickle.HIGHEST_PROTOCOL)
    elif fs_type == 'torch':
        buf = io.BytesIO()
        torch.save(data, buf)
        payload = buf.getvalue()
    elif fs_type == 'pyarrow':
        # kept for API parity; pyarrow's generic serializer was removed
        # upstream, numpy-pickle is the equivalent fast path
        payload = pickle.dumps(_tensor_to_np(data), protocol=pickle.HIGHEST_PROTOCOL)
    else:
        raise KeyError(fs_type)
    if compress:
        return _MAGIC_ZLIB + _compress(payload, level)
    return _MAGIC_RAW + payload


def loads(blob, fs_type='nppickle'):
    magic, payload = blob[:4], blob[4:]
    if magic == _MAGIC_ZLIB:
        payload = _decompress(payload)
    elif magic != _MAGIC_RAW:
        payload = blob      # uncompressed legacy payload
    if fs_type in ('pickle', 'cPickle'):
        return pickle.loads(payload)
    if fs_type in ('nppickle', 'pyarrow'):
        return _np_to_tensor(pickle.loads(payload))
    if fs_type == 'torch':
        return torch.load(io.BytesIO(payload), map_location='cpu', weights_only=False)
    raise KeyError(fs_type)


def read_file(path, fs_type='torch'):
    if fs_type == 'torch':
        return torch.load(path, map_location='cpu', weights_only=False)
    with open(path, 'rb') as f:
        return loads(f.read(), fs_type=fs_type)


def save_file(path, data, fs_type='torch'):
    if fs_type == 'torch':
        torch.save(data, path)
        return
    with open(path, 'wb') as f:
        f.write(dumps(data, fs_type=fs_type))
