#!/usr/bin/env python3
"""Bounded end-to-end league soak: the REAL `bin/rl_train` single-node
stack (coordinator + league HTTP control plane + RL learner + actor with
batched GPU inference over mock SC2 envs) run for a few minutes on one
GPU, then shut down cleanly.

This is the reference's whole-system smoke (BASELINE config 5 wiring) as
an on-hardware artifact: league job assignment over HTTP, actor rollouts
with teacher logits, trajectory transport through the coordinator/adapter,
learner V-trace iterations with the HIP kernel path, checkpoint save.

Usage: python tools/rl_soak.py [--seconds 360] [--iters 10] [--out DIR]
"""
import argparse
import os
import signal
import subprocess
import sys
import tempfile
import time

import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def merge(d, o):
    for k, v in o.items():
        if isinstance(v, dict) and isinstance(d.get(k), dict):
            merge(d[k], v)
        else:
            d[k] = v


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--seconds', type=int, default=360)
    p.add_argument('--iters', type=int, default=10)
    p.add_argument('--cpu', action='store_true')
    p.add_argument('--out', default=os.path.join(REPO, 'gpurun_out'))
    args = p.parse_args()
    os.makedirs(args.out, exist_ok=True)

    base = yaml.safe_load(
        open(os.path.join(REPO, 'distar_amd/bin/rl_user_config.yaml')))
    merge(base, {
        'common': {'experiment_name': 'rl_soak'},
        'actor': {'env_num': 1, 'traj_len': 8, 'episode_num': 2,
                  'use_cuda': not args.cpu},
        'learner': {'use_cuda': None if not args.cpu else False,
                    'value_pretrain_iters': 0,
                    'data': {'buffer_size': 4, 'batch_size': 2,
                             'trajectory_length': 8, 'num_workers': 1,
                             'use_async_cuda': not args.cpu},
                    'hook': {'after_iter': {
                        'log_show': {'priority': 30, 'ext_args': {'freq': 1}},
                        'save_ckpt': {'ext_args': {'freq': 5}}}}},
    })
    # the single-node spawner picks fresh control-plane ports
    base['communication'].pop('coordinator_port', None)
    base['communication'].pop('league_port', None)
    cfgf = tempfile.NamedTemporaryFile('w', suffix='.yaml', delete=False)
    yaml.safe_dump(base, cfgf)
    cfgf.close()

    logp = os.path.join(args.out, 'rl_soak.log')
    t0 = time.time()
    with open(logp, 'w') as log:
        proc = subprocess.Popen(
            [sys.executable, '-m', 'distar_amd.bin.rl_train',
             '--config', cfgf.name, '--max-iterations', str(args.iters)],
            stdout=log, stderr=subprocess.STDOUT, cwd=REPO,
            start_new_session=True)
        try:
            rc = proc.wait(timeout=args.seconds)
        except subprocess.TimeoutExpired:
            # SIGINT -> KeyboardInterrupt -> multiprocessing cleans up the
            # daemon coordinator/league/learner children
            os.killpg(proc.pid, signal.SIGINT)
            try:
                rc = proc.wait(timeout=20)
            except subprocess.TimeoutExpired:
                os.killpg(proc.pid, signal.SIGKILL)
                rc = proc.wait()
    dur = time.time() - t0
    print(f'rl_train rc={rc} after {dur:.0f}s; log: {logp}')
    tail = open(logp).read()[-2500:]
    print(tail)
    # experiment artifacts (learner log, checkpoints)
    for root, _dirs, files in os.walk(os.path.join(REPO, 'experiments')):
        for f in files:
            fp = os.path.join(root, f)
            print(f'{os.path.getsize(fp):>12}  {os.path.relpath(fp, REPO)}')


if __name__ == '__main__':
    main()
