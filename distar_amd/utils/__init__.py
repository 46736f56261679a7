from .config import Config, read_config, save_config, deep_merge_dicts
