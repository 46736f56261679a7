"""Reinforcement-learning entry point (reference `distar/bin/rl_train.py`):

  python -m distar_amd.bin.rl_train --config <yaml> [--type learner|actor|
      league|coordinator] [--task-id N]

Without --type, spawns coordinator + league + learner processes and runs the
actor in the main process (the reference's single-node mode,
`bin/rl_train.py:144-154`).  With --type, runs a single role (multi-node
deployments launch each role separately and point `communication.*` at the
shared coordinator/league addresses).
"""
import argparse
import multiprocessing as mp
import os
import time

from ..utils.config import Config, deep_merge_dicts, read_config
from ..utils.http import pick_unused_port

DEFAULT_RL_CONFIG = os.path.join(os.path.dirname(__file__), 'rl_user_config.yaml')


def get_args(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument('--config', default=DEFAULT_RL_CONFIG)
    p.add_argument('--type', default=None,
                   choices=[None, 'learner', 'actor', 'league', 'coordinator'])
    p.add_argument('--player-id', default='MP0')
    p.add_argument('--init-method', dest='init_method', default=None)
    p.add_argument('--rank', type=int, default=0)
    p.add_argument('--world-size', dest='world_size', type=int, default=1)
    p.add_argument('--max-iterations', type=int, default=None)
    return p.parse_args(argv)


def load_cfg(args):
    cfg = read_config(args.config) if os.path.exists(args.config) else Config({})
    return cfg


def league_run(cfg, port_value):
    from ..league.league import League
    from ..league.api import create_league_server
    league = League(cfg)
    server = create_league_server(league, port=cfg.get('communication', {}).get('league_port'))
    port_value.value = server.port
    server.start(daemon=False)
    print(f'league api on port {server.port}')
    while True:
        time.sleep(60)


def coordinator_run(cfg, port_value):
    from ..data.coordinator import Coordinator
    coord = Coordinator(cfg)
    port_value.value = coord.port
    coord.run(daemon=False)
    print(f'coordinator on port {coord.port}')
    while True:
        time.sleep(60)


def learner_run(cfg, args):
    from ..learner.rl_learner import RLLearner
    from ..actor.comm import LearnerComm
    cfg = deep_merge_dicts(cfg, {'learner': {'player_id': args.player_id}})
    if args.world_size > 1 or 'RANK' in os.environ:
        cfg = deep_merge_dicts(cfg, {'learner': {'use_distributed': True}})
        learner = RLLearner(cfg, method='torch', init_method=args.init_method,
                            rank=args.rank, world_size=args.world_size)
    else:
        learner = RLLearner(cfg)
    if cfg.get('communication', {}).get('league_port'):
        comm = LearnerComm(cfg)
        comm.register_learner(learner)
        learner._setup_comm_hooks(comm)
    if getattr(learner, '_rank', 0) == 0:
        # live-control endpoints (reference rl_train.py:19-51 serves a Flask
        # debug app on a picked port); port logged to the learner log
        learner.start_debug_server()
    learner.run(max_iterations=args.max_iterations)


def actor_run(cfg):
    from ..actor.actor import Actor
    Actor(cfg).run()


def main(argv=None):
    args = get_args(argv)
    cfg = load_cfg(args)
    if args.type == 'league':
        league_run(cfg, mp.Value('i', 0))
    elif args.type == 'coordinator':
        coordinator_run(cfg, mp.Value('i', 0))
    elif args.type == 'learner':
        learner_run(cfg, args)
    elif args.type == 'actor':
        actor_run(cfg)
    else:
        # single-node mode: spawn control plane + learner, actor in main proc
        ctx = mp.get_context('spawn')
        coord_port = ctx.Value('i', 0)
        league_port = ctx.Value('i', 0)
        procs = [ctx.Process(target=coordinator_run, args=(cfg, coord_port), daemon=True),
                 ctx.Process(target=league_run, args=(cfg, league_port), daemon=True)]
        for p in procs:
            p.start()
        while coord_port.value == 0 or league_port.value == 0:
            time.sleep(0.2)
        cfg = deep_merge_dicts(cfg, {'communication': {
            'coordinator_ip': '127.0.0.1', 'coordinator_port': coord_port.value,
            'league_ip': '127.0.0.1', 'league_port': league_port.value}})
        learner_proc = ctx.Process(target=learner_run, args=(cfg, args), daemon=True)
        learner_proc.start()
        actor_run(cfg)


if __name__ == '__main__':
    main()
