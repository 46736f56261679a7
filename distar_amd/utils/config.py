"""Layered YAML config system.

Mirrors the reference's behavior (`distar/ctools/utils/config_helper.py:10-113`):
module-default YAML deep-merged with a user YAML, attribute-style access.
Implementation is our own (the reference depends on the external `easydict`
package, which this image does not ship).
"""
import copy
import os

import yaml


class Config(dict):
    """Attribute-accessible dict. Nested dicts are converted recursively."""

    def __init__(self, d=None, **kwargs):
        super().__init__()
        d = dict(d or {}, **kwargs)
        for k, v in d.items():
            self[k] = v

    @staticmethod
    def _wrap(value):
        if isinstance(value, dict) and not isinstance(value, Config):
            return Config(value)
        if isinstance(value, (list, tuple)):
            return type(value)(Config._wrap(x) for x in value)
        return value

    def __setitem__(self, key, value):
        super().__setitem__(key, Config._wrap(value))

    def __setattr__(self, key, value):
        self[key] = value

    def __getattr__(self, key):
        try:
            return self[key]
        except KeyError:
            raise AttributeError(key)

    def __delattr__(self, key):
        try:
            del self[key]
        except KeyError:
            raise AttributeError(key)

    def update(self, other=None, **kwargs):
        for k, v in dict(other or {}, **kwargs).items():
            self[k] = v

    def __deepcopy__(self, memo):
        return Config({k: copy.deepcopy(dict(v) if isinstance(v, Config) else v, memo)
                       for k, v in self.items()})

    def to_dict(self):
        out = {}
        for k, v in self.items():
            if isinstance(v, Config):
                out[k] = v.to_dict()
            elif isinstance(v, (list, tuple)):
                out[k] = type(v)(x.to_dict() if isinstance(x, Config) else x for x in v)
            else:
                out[k] = v
        return out


def read_config(path):
    """YAML file -> Config (reference: config_helper.read_config)."""
    if not os.path.exists(path):
        raise FileNotFoundError(path)
    with open(path, 'r') as f:
        data = yaml.safe_load(f)
    return Config(data or {})


def save_config(config, path):
    with open(path, 'w') as f:
        yaml.safe_dump(config.to_dict() if isinstance(config, Config) else config, f,
                       default_flow_style=False)


def deep_merge_dicts(original, new_dict):
    """Recursively merge ``new_dict`` over ``original`` without mutating either
    (reference: config_helper.deep_merge_dicts/deep_update)."""
    original = original if original is not None else {}
    new_dict = new_dict if new_dict is not None else {}
    merged = Config(copy.deepcopy(dict(original)) if original else {})
    for key, value in new_dict.items():
        if key in merged and isinstance(merged[key], dict) and isinstance(value, dict):
            merged[key] = deep_merge_dicts(merged[key], value)
        else:
            merged[key] = copy.deepcopy(value) if isinstance(value, dict) else value
    return merged
