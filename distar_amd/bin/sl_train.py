"""Supervised-learning entry point (reference `distar/bin/sl_train.py`):

  python -m distar_amd.bin.sl_train --config <yaml> [--type learner|
      coordinator|replay_actor] [--init-method tcp://...] [--rank R]
      [--world-size W]

Roles: 'learner' runs the SL learner (distributed when --world-size > 1,
RCCL on GPU / gloo on CPU); 'coordinator' runs the Adapter metadata broker;
'replay_actor' runs a replay-decoding fleet node.
"""
import argparse
import os

from ..utils.config import Config, read_config, deep_merge_dicts

DEFAULT_SL_CONFIG = os.path.join(os.path.dirname(__file__), 'sl_user_config.yaml')


def get_args(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument('--config', default=DEFAULT_SL_CONFIG)
    p.add_argument('--type', default='learner',
                   choices=['learner', 'coordinator', 'replay_actor'])
    p.add_argument('--init-method', '--init_method', dest='init_method', default=None)
    p.add_argument('--rank', type=int, default=0)
    p.add_argument('--world-size', '--world_size', dest='world_size', type=int, default=1)
    p.add_argument('--max-iterations', type=int, default=None)
    return p.parse_args(argv)


def main(argv=None):
    args = get_args(argv)
    cfg = read_config(args.config) if os.path.exists(args.config) else Config({})
    if args.type == 'learner':
        from ..learner.sl_learner import SLLearner
        if args.world_size > 1 or 'RANK' in os.environ:
            cfg = deep_merge_dicts(cfg, {'learner': {'use_distributed': True}})
            learner = SLLearner(cfg, method='torch', init_method=args.init_method,
                                rank=args.rank, world_size=args.world_size)
        else:
            learner = SLLearner(cfg)
        learner.run(max_iterations=args.max_iterations)
    elif args.type == 'coordinator':
        import time
        from ..data.coordinator import Coordinator
        coord = Coordinator(cfg).run(daemon=False)
        print(f'coordinator on port {coord.port}')
        while True:
            time.sleep(60)
    elif args.type == 'replay_actor':
        from ..data.replay_actor import ReplayActor
        ReplayActor(cfg).run()


if __name__ == '__main__':
    main()
