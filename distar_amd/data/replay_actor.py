"""Remote replay-decoding fleet (reference
`ctools/worker/actor/replay_actor.py:10-77`): shard replay paths over
SLURM-style task ids x worker processes; each worker decodes full replays
and `Adapter.push`es the step lists for remote SL dataloaders to pull."""
import os

import torch.multiprocessing as mp

from .adapter import Adapter


def _worker(cfg, paths, worker_index):
    from .replay_decoder import ReplayDecoder
    adapter = Adapter(cfg)
    decoder = ReplayDecoder(cfg)
    for path in paths:
        for player_idx in range(2):
            data = decoder.run(path, player_idx)
            if data:
                adapter.push(data, token='replay', fs_type='nppickle')
    decoder.close()


class ReplayActor:
    def __init__(self, cfg):
        self._whole_cfg = cfg
        data_cfg = cfg.learner.data
        paths = []
        src = data_cfg.train_data_file
        if os.path.isfile(src):
            with open(src) as f:
                paths = [l.strip() for l in f if l.strip()]
        elif os.path.isdir(src):
            paths = [os.path.join(src, p) for p in sorted(os.listdir(src))]
        task_id = int(os.environ.get('SLURM_PROCID', 0))
        n_tasks = int(os.environ.get('SLURM_NTASKS', 1))
        paths = paths[task_id::n_tasks]
        self._paths = paths
        self._num_workers = data_cfg.get('num_replay_workers', 2)

    def run(self):
        procs = []
        for i in range(self._num_workers):
            shard = self._paths[i::self._num_workers]
            p = mp.Process(target=_worker, args=(self._whole_cfg, shard, i),
                           daemon=True)
            p.start()
            procs.append(p)
        for p in procs:
            p.join()
