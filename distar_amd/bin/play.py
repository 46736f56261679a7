"""Play/eval entry point (reference `distar/bin/play.py:27-87`):
human_vs_agent / agent_vs_agent / agent_vs_bot with `job_type='eval_test'`,
realtime mode and the per-race action legality mask enabled
(`common.type='play'`).

Real games need StarCraft II + s2clientprotocol (gated); `--env mock` runs
the same code path on the synthetic environment for harness checks.
"""
import argparse
import os

from ..utils.config import Config, deep_merge_dicts, read_config

DEFAULT_PLAY_CONFIG = os.path.join(os.path.dirname(__file__), 'user_config.yaml')


def get_args(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument('--config', default=DEFAULT_PLAY_CONFIG)
    p.add_argument('--game-type', default='agent_vs_agent',
                   choices=['human_vs_agent', 'agent_vs_agent', 'agent_vs_bot'])
    p.add_argument('--model1', default='rl_model')
    p.add_argument('--model2', default='rl_model')
    p.add_argument('--env', default='sc2', choices=['sc2', 'mock'])
    p.add_argument('--episodes', type=int, default=1)
    return p.parse_args(argv)


def main(argv=None):
    args = get_args(argv)
    cfg = read_config(args.config) if os.path.exists(args.config) else Config({})
    overrides = {
        'common': {'type': 'play'},
        'actor': {'job_type': 'eval_test', 'env_type': args.env,
                  'episode_num': args.episodes},
        'env': {'realtime': args.env == 'sc2'},
    }
    cfg = deep_merge_dicts(cfg, overrides)
    model_dir = os.path.join(os.path.dirname(__file__))
    for i, model in enumerate([args.model1, args.model2]):
        path = model if os.path.exists(model) else \
            os.path.join(model_dir, f'{model}.pth')
        cfg = deep_merge_dicts(cfg, {'actor': {f'model{i}_path': path}})
    from ..actor.actor import Actor
    actor = Actor(cfg)
    results = actor.run()
    for r in results:
        print({k: v.get('winloss') if isinstance(v, dict) else v
               for k, v in r.items()})


if __name__ == '__main__':
    main()
