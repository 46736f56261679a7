#!/usr/bin/env python3
"""One SL/RL learner step under torch.profiler: top ops by device time with
input shapes — attributes copy/elementwise tails to their source ops
(rocprof names every D2D copy `__amd_rocclr_copyBuffer`, which hides the
caller)."""
import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--mode', choices=['sl', 'rl'], default='sl')
    p.add_argument('--rows', type=int, default=40)
    p.add_argument('--stack', action='store_true',
                   help='collect python stacks and group by source location '
                        '(attributes copy_/elementwise tails to model code)')
    p.add_argument('--filter', default=None,
                   help='only print rows whose name contains this substring')
    args = p.parse_args()
    import bench
    ns = argparse.Namespace(batch=32 if args.mode == 'sl' else 16, traj=64,
                            entities=256, pool=1, bucket_mb=64, no_amp=False,
                            mode=args.mode, no_value_feature=False)
    device = torch.device('cuda')
    cls = bench.SLBench if args.mode == 'sl' else bench.RLBench
    b = cls(ns, device, True)
    for i in range(2):
        b.step(i)
    torch.cuda.synchronize()
    from torch.profiler import ProfilerActivity, profile
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=True, with_stack=args.stack) as prof:
        b.step(2)
        torch.cuda.synchronize()
    ka = prof.key_averages(group_by_input_shape=True,
                           group_by_stack_n=12 if args.stack else 0)
    if args.filter:
        rows = [e for e in ka if args.filter in e.key]
        for e in sorted(rows, key=lambda e: -e.self_device_time_total):
            print(f'{e.key}  shapes={e.input_shapes}  '
                  f'self_cuda={e.self_device_time_total / 1000:.3f}ms  '
                  f'calls={e.count}')
            for ln in (e.stack or [])[:12]:
                print('    ', ln)
    else:
        print(ka.table(sort_by='self_cuda_time_total', row_limit=args.rows,
                       max_src_column_width=60))


if __name__ == '__main__':
    main()
