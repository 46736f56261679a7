"""Data-plane serialization microbench (CPU).

Measures dumps/loads throughput of the trajectory transport codecs
(`distar_amd/utils/serialize.py`) on a realistic RL trajectory payload
(traj_len=16 rollout steps, the actor->learner unit of transfer), plus the
native GIL-released zlib codec vs the stdlib zlib module.

Run: python tools/bench_serialize.py [--iters N]  -> prints a table; copy the
output into profiles/ for the record.
"""
import argparse
import random
import time
import zlib

import torch

from distar_amd.lib.fake_data import fake_rl_step
from distar_amd.utils import serialize


def payload_traj(traj_len=16):
    rng = random.Random(0)
    torch.manual_seed(0)
    return [fake_rl_step(256, rng) for _ in range(traj_len)]


def timeit(fn, iters):
    fn()                                    # warm
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--iters', type=int, default=10)
    ap.add_argument('--traj-len', type=int, default=16)
    args = ap.parse_args()

    traj = payload_traj(args.traj_len)
    raw = serialize.dumps(traj, fs_type='nppickle', compress=False)
    size_mb = len(raw) / 1e6
    print(f'payload: {args.traj_len}-step RL trajectory, {size_mb:.1f} MB raw pickle')
    rows = []
    for fs_type in ('nppickle', 'pickle', 'torch'):
        for compress in (False, True):
            blob = serialize.dumps(traj, fs_type=fs_type, compress=compress)
            td = timeit(lambda: serialize.dumps(traj, fs_type=fs_type,
                                                compress=compress), args.iters)
            tl = timeit(lambda: serialize.loads(blob, fs_type=fs_type), args.iters)
            rows.append((f'{fs_type}{"+z" if compress else ""}', len(blob) / 1e6,
                         td * 1e3, size_mb / td, tl * 1e3, size_mb / tl))
    print(f'{"codec":>12s} {"MB":>7s} {"dump ms":>9s} {"dump MB/s":>10s} '
          f'{"load ms":>9s} {"load MB/s":>10s}')
    for name, mb, dms, dmbs, lms, lmbs in rows:
        print(f'{name:>12s} {mb:7.1f} {dms:9.2f} {dmbs:10.0f} {lms:9.2f} {lmbs:10.0f}')

    # native codec vs stdlib zlib on the pickled payload
    import inspect
    native = '_native_codec' in inspect.getsource(serialize._compress)
    print(f'\nzlib level-1 on the raw pickle (native codec loaded: {native}):')
    for name, comp, decomp in (
            ('stdlib', lambda: zlib.compress(raw, 1),
             lambda b: zlib.decompress(b)),
            ('serialize._compress', lambda: serialize._compress(raw, 1),
             lambda b: serialize._decompress(b))):
        blob = comp()
        tc = timeit(comp, args.iters)
        tdc = timeit(lambda: decomp(blob), args.iters)
        print(f'{name:>20s}: compress {size_mb/tc:6.0f} MB/s  '
              f'decompress {size_mb/tdc:6.0f} MB/s  ratio {len(raw)/len(blob):.2f}x')

    # GIL-release scaling: 4 concurrent decompressions (the learner-side
    # pattern: adapter pull worker threads); native releases the GIL, stdlib
    # zlib also releases it for large buffers -- both should scale
    import threading
    blob = serialize._compress(raw, 1)
    for name, fn in (('stdlib', lambda: zlib.decompress(blob)),
                     ('native', lambda: serialize._decompress(blob))):
        t1 = timeit(fn, args.iters)
        def four():
            ts = [threading.Thread(target=fn) for _ in range(4)]
            for t in ts:
                t.start()
            for t in ts:
                t.join()
        t4 = timeit(four, max(1, args.iters // 2))
        print(f'{name:>20s}: 1-thread {size_mb/t1:6.0f} MB/s, '
              f'4-thread aggregate {4*size_mb/t4:6.0f} MB/s '
              f'({t4/t1:.2f}x wall of 1-thread)')


if __name__ == '__main__':
    main()
