#!/usr/bin/env python3
"""Flagship learner benchmark (driver contract — see BASELINE.json).

`python bench.py --gpus N --steps K --warmup W [--mode both|sl|rl]` runs the
SL and/or RL learner train step on synthetic data of the reference's
shapes with random-init weights, W untimed warmup steps per mode, then times
exactly K steps bracketed by barrier + synchronize, takes the MAX step time
over ranks, and rank 0 prints ONE JSON line.

BASELINE.json's metric has two halves — "learner samples/sec (SL) +
trajectory-steps/sec (RL V-trace)" — so the default mode measures BOTH:
the headline value/ms_per_step are the SL numbers and the RL half is
reported under config.rl (value, ms_per_step, vs_baseline), each timed in
its own warmup+K-step bracketed window.

SL config: batch=32 x seq_len=64 per GPU, bf16, 512-entity observations
(the reference SL slab always runs the transformer at the full 512-entity
width).  RL config: batch=16 x traj 64 per GPU with the full value-feature
critic path.  Baseline anchors: 384 SL samples/s per A100 (21,504
samples/iter at ~1 s on 56xA100 —
docs/guidance_to_small_scale_training.md:178-184); 256 RL
trajectory-steps/s per A100 (8.2k steps/s per 32-A100 learner, :280-284).
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from distar_amd.lib.fake_data import (fake_rl_learner_data_fast,           # noqa: E402
                                      fake_sl_batch_fast)
from distar_amd.losses import ReinforcementLoss, SupervisedLoss            # noqa: E402
from distar_amd.models import Model                                        # noqa: E402
from distar_amd.parallel.ddp import DistModule                             # noqa: E402
from distar_amd.parallel.dist import barrier, dist_init, get_rank, get_world_size  # noqa: E402
from distar_amd.utils.config import Config                                 # noqa: E402
from distar_amd.utils.data import to_device                                # noqa: E402
from distar_amd.utils.grad_clip import build_grad_clip                     # noqa: E402

SL_BASELINE_PER_GPU = 384.0     # samples/s per A100
RL_BASELINE_PER_GPU = 256.0     # trajectory-steps/s per A100


class SLBench:
    def __init__(self, args, device, use_amp):
        self.args = args
        self.device = device
        self.use_amp = use_amp
        torch.manual_seed(1234 + get_rank())
        self.model = Model(Config({'common': {'type': 'train'}}), temperature=1.0)
        self.model = self.model.to(device)
        if os.environ.get('DISTAR_AMD_CHANNELS_LAST') == '1':
            self.model = self.model.to(memory_format=torch.channels_last)
        if get_world_size() > 1:
            self.model = DistModule(self.model, bucket_cap_mb=args.bucket_mb)
        self.loss = SupervisedLoss(Config({'learner': {}}))
        self.grad_clip = build_grad_clip(Config({'type': 'momentum_norm', 'threshold': 1.0}))
        self.optimizer = torch.optim.Adam(self.model.parameters(), lr=1e-3,
                                          fused=torch.cuda.is_available())
        B, T = args.batch, args.traj
        self.batches = [
            to_device(fake_sl_batch_fast(B, T, seed=100 * get_rank() + i), device)
            for i in range(args.pool)
        ]
        H = self.model.cfg.encoder.core_lstm.hidden_size if get_world_size() == 1 \
            else self.model.module.cfg.encoder.core_lstm.hidden_size
        z = torch.zeros(B, H, device=device)
        self.hidden = [(z, z) for _ in range(3)]
        self.samples_per_step = B * T

    def step(self, i):
        data = self.batches[i % len(self.batches)]
        model = self.model.module if isinstance(self.model, DistModule) else self.model
        with torch.autocast('cuda', dtype=torch.bfloat16, enabled=self.use_amp):
            logits, infer_action, hidden = model.sl_train(
                spatial_info=data['spatial_info'], scalar_info=data['scalar_info'],
                entity_info=data['entity_info'], entity_num=data['entity_num'],
                selected_units_num=data['selected_units_num'],
                traj_lens=data['traj_lens'], hidden_state=self.hidden,
                action_info=data['action_info'])
            ld = self.loss.compute_loss(logits, data['action_info'], data['action_mask'],
                                        data['selected_units_num'], data['entity_num'],
                                        infer_action)
        self.optimizer.zero_grad(set_to_none=True)
        ld['total_loss'].backward()
        if isinstance(self.model, DistModule):
            self.model.sync_gradients()
        self.grad_clip.apply(self.model.parameters())
        self.optimizer.step()
        self.hidden = [(h.detach(), c.detach()) for h, c in hidden]


class RLBench:
    def __init__(self, args, device, use_amp):
        self.args = args
        self.device = device
        self.use_amp = use_amp
        torch.manual_seed(1234 + get_rank())
        self.model = Model(
            Config({'common': {'type': 'train'},
                    'learner': {'use_value_feature': not args.no_value_feature},
                    'model': {'enable_baselines':
                              ['winloss', 'build_order', 'built_unit', 'battle']}}),
            use_value_network=True).to(device)
        if get_world_size() > 1:
            self.model = DistModule(self.model, bucket_cap_mb=args.bucket_mb)
        self.loss = ReinforcementLoss(Config({}), 'MP0')
        self.grad_clip = build_grad_clip(Config({'type': 'pytorch_norm', 'threshold': 1.0}))
        self.optimizer = torch.optim.Adam(self.model.parameters(), lr=1e-5,
                                          betas=(0.0, 0.99), eps=1e-5,
                                          fused=torch.cuda.is_available())
        batches = []
        for i in range(args.pool):
            d = fake_rl_learner_data_fast(args.batch, args.traj,
                                          entity_num=args.entities,
                                          seed=100 * get_rank() + i,
                                          value_feature=not args.no_value_feature)
            d.pop('model_last_iter')
            batches.append(to_device(d, device))
        self.batches = batches
        self.samples_per_step = args.batch * args.traj

    def step(self, i):
        data = self.batches[i % len(self.batches)]
        model = self.model.module if isinstance(self.model, DistModule) else self.model
        with torch.autocast('cuda', dtype=torch.bfloat16, enabled=self.use_amp):
            out = model.rl_learner_forward(**data)
            ld = self.loss.compute_loss(out)
        self.optimizer.zero_grad(set_to_none=True)
        ld['total_loss'].backward()
        if isinstance(self.model, DistModule):
            self.model.sync_gradients()
        self.grad_clip.apply(self.model.parameters())
        self.optimizer.step()


def _timed(bench, steps, warmup, has_gpu, n_gpus):
    for i in range(warmup):
        bench.step(i)
    if has_gpu:
        torch.cuda.synchronize()
    barrier()
    t0 = time.perf_counter()
    for i in range(steps):
        bench.step(warmup + i)
    if has_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    barrier()
    t = torch.tensor([elapsed], device='cuda' if has_gpu else 'cpu')
    if n_gpus > 1:
        import torch.distributed as dist
        dist.all_reduce(t, op=dist.ReduceOp.MAX)   # max step time over ranks
    return float(t[0])


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--gpus', type=int, default=1)
    p.add_argument('--steps', type=int, default=8)
    p.add_argument('--warmup', type=int, default=3)
    p.add_argument('--mode', choices=['both', 'sl', 'rl'], default='both')
    p.add_argument('--batch', type=int, default=None)
    p.add_argument('--traj', type=int, default=None)
    p.add_argument('--entities', type=int, default=256)
    p.add_argument('--pool', type=int, default=2)
    p.add_argument('--bucket-mb', type=int, default=64)
    p.add_argument('--no-amp', action='store_true')
    p.add_argument('--no-value-feature', action='store_true',
                   help='RL mode: drop the opponent-side value features '
                        '(the reference RL config trains WITH them)')
    args = p.parse_args()
    if args.traj is None:
        args.traj = 64

    world = int(os.environ.get('WORLD_SIZE', '1'))
    if world > 1:
        dist_init(method='torch')
    rank = get_rank()
    n_gpus = get_world_size()
    has_gpu = torch.cuda.is_available()
    device = torch.device('cuda', int(os.environ.get('LOCAL_RANK', 0))) if has_gpu \
        else torch.device('cpu')
    if has_gpu:
        torch.cuda.set_device(device)
        # NOTE: cudnn.benchmark=True (MIOpen exhaustive find) costs 10+ min
        # of solver compilation on a fresh box per unique conv shape —
        # measured r2: it blew an 840 s bench window.  Immediate mode keeps
        # the r01 conv behavior; the real fix is the hand-written K4 conv.
    use_amp = has_gpu and not args.no_amp

    import copy
    results = {}
    modes = ['sl', 'rl'] if args.mode == 'both' else [args.mode]
    for mode in modes:
        margs = copy.copy(args)
        margs.mode = mode
        margs.batch = args.batch if args.batch is not None \
            else (32 if mode == 'sl' else 16)
        bench = (SLBench if mode == 'sl' else RLBench)(margs, device, use_amp)
        elapsed = _timed(bench, args.steps, args.warmup, has_gpu, n_gpus)
        per_gpu = bench.samples_per_step / (elapsed / args.steps)
        baseline = SL_BASELINE_PER_GPU if mode == 'sl' else RL_BASELINE_PER_GPU
        results[mode] = {
            'value': round(per_gpu * n_gpus, 1),
            'ms_per_step': round(elapsed / args.steps * 1000, 2),
            'vs_baseline': round(per_gpu / baseline, 3),
            'global_batch': margs.batch * n_gpus,
        }
        del bench
        if has_gpu:
            torch.cuda.empty_cache()

    head = results.get('sl') or results['rl']
    head_mode = 'sl' if 'sl' in results else 'rl'
    if rank == 0:
        result = {
            'metric': 'SL learner samples/s + RL trajectory-steps/s'
                      if args.mode == 'both' else
                      ('SL learner samples/s' if head_mode == 'sl'
                       else 'RL learner trajectory-steps/s'),
            'value': head['value'],
            'unit': 'samples/s' if head_mode == 'sl' else 'steps/s',
            'n_gpus': n_gpus, 'steps': args.steps, 'warmup': args.warmup,
            'ms_per_step': head['ms_per_step'],
            'higher_is_better': True, 'scaling': 'weak',
            'vs_baseline': head['vs_baseline'],
            'dtype': 'bf16' if use_amp else 'fp32',
            'data': 'synthetic',
            'config': {
                'model': 'alphastar-zerg (reference default dims)',
                'global_batch': head['global_batch'], 'seq_len': args.traj,
                'entities': 512 if head_mode == 'sl' else args.entities,
                'parallelism': f'dp{n_gpus}',
                'mode': args.mode,
                'value_feature': not args.no_value_feature,
            },
        }
        if args.mode == 'both':
            result['config']['rl'] = {
                **results['rl'],
                'entities': args.entities,
                'unit': 'trajectory-steps/s',
                'baseline_per_gpu': RL_BASELINE_PER_GPU,
            }
        print(json.dumps(result))


if __name__ == '__main__':
    main()
