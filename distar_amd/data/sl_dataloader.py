"""Supervised-learning dataloader: shared-memory batch lanes fed by worker
processes.

Functional parity with the reference's
`sl_training/sl_dataloader.py:19-190`: a pre-allocated shared-memory batch
of (batch_size x trajectory_length) step slots; each worker owns one replay
at a time and copies successive trajectory_length windows into its assigned
batch lane (entity/SU/spatial zero-padded), round-tripping lane tokens over
pipes; `__next__` returns the shared batch + traj_lens/new_episodes.

Data sources per worker:
  - 'replay'  decode .SC2Replay files live (ReplayDecoder, needs SC2),
  - 'remote'  Adapter.pull from a replay-actor fleet,
  - 'offline' pre-decoded trajectory files (torch-saved step lists) — the
    source usable in this offline image and in tests.
"""
import os
import random
import time

import torch
import torch.multiprocessing as mp

from ..lib.consts import fake_step_data
from ..parallel.dist import get_rank, get_world_size


def send_data(worker_queue, pipe, worker_index, data, shared_step_data,
              trajectory_length):
    """Copy one replay's successive windows into assigned lanes (reference
    sl_dataloader.py:19-68)."""
    worker_queue.put(worker_index)
    start = 0
    replay_length = len(data)
    while True:
        if pipe.poll(0.002):
            batch_index = pipe.recv()
            end = min(start + trajectory_length, replay_length)
            for i in range(start, end):
                step_data = data[i]
                data_idx = batch_index * trajectory_length + i - start
                entity_num = step_data['entity_num']
                selected_units_num = step_data['selected_units_num']
                for k, v in step_data.items():
                    if isinstance(v, torch.Tensor):
                        shared_step_data[k][data_idx].copy_(v)
                    elif isinstance(v, dict):
                        for _k, _v in v.items():
                            if _k not in shared_step_data[k]:
                                continue
                            dst = shared_step_data[k][_k]
                            if k == 'action_info' and _k == 'selected_units':
                                if selected_units_num > 0:
                                    dst[data_idx, :selected_units_num].copy_(
                                        _v[:selected_units_num])
                            elif k == 'entity_info':
                                dst[data_idx, :entity_num].copy_(_v[:entity_num])
                            elif k == 'spatial_info' and 'effect' not in _k:
                                h, w = _v.shape
                                dst[data_idx] *= 0
                                dst[data_idx, :h, :w].copy_(_v)
                            else:
                                dst[data_idx].copy_(_v)
            for i in range(end, start + trajectory_length):
                data_idx = batch_index * trajectory_length + i - start
                for k in shared_step_data['action_mask']:
                    shared_step_data['action_mask'][k][data_idx].copy_(
                        torch.tensor(0).bool())
            new_episode = start == 0
            end_episode = end == replay_length
            pipe.send((new_episode, end - start, end_episode))
            start = end
            if end_episode:
                return


def worker_loop(cfg, paths, pipe, shared_step_data, worker_queue, worker_index):
    torch.set_num_threads(1)
    data_cfg = cfg.learner.data
    source = data_cfg.get('source', 'offline')
    traj_len = data_cfg.trajectory_length
    if source == 'remote':
        from .adapter import Adapter
        adapter = Adapter(cfg)
        while True:
            pulled = adapter.pull(fs_type='nppickle', sleep_time=0.2,
                                  size=1, token='replay')
            if pulled:
                send_data(worker_queue, pipe, worker_index, pulled[0],
                          shared_step_data, traj_len)
    elif source == 'replay':
        from .replay_decoder import ReplayDecoder
        decoder = ReplayDecoder(cfg)
        data_idx, player_idx = 0, 0
        while data_idx < len(paths):
            data = decoder.run(paths[data_idx], player_idx)
            player_idx = (player_idx + 1) % 2
            if player_idx == 0:
                data_idx += 1
            if data is not None:
                send_data(worker_queue, pipe, worker_index, data,
                          shared_step_data, traj_len)
        print('[SLDataloader] ran out of replays, training data done')
    else:   # offline: torch-saved step lists
        idx = 0
        while True:
            path = paths[idx % len(paths)]
            idx += 1
            try:
                data = torch.load(path, map_location='cpu', weights_only=False)
            except Exception as e:  # noqa: BLE001
                print(f'[SLDataloader] bad file {path}: {e!r}')
                continue
            send_data(worker_queue, pipe, worker_index, data,
                      shared_step_data, traj_len)


class SLDataloader:
    def __init__(self, cfg):
        torch.set_num_threads(1)
        self.use_cuda = cfg.learner.use_cuda and torch.cuda.is_available()
        self.device = torch.cuda.current_device() if self.use_cuda else None
        self.cfg = cfg.learner.data
        self.batch_size = self.cfg.batch_size
        self.trajectory_length = self.cfg.trajectory_length
        data_paths = []
        train_file = self.cfg.train_data_file
        if os.path.isfile(train_file):
            with open(train_file) as f:
                data_paths = [l.strip() for l in f if l.strip()]
        elif os.path.isdir(train_file):
            data_paths = [os.path.join(train_file, p)
                          for p in sorted(os.listdir(train_file))]
        data_paths = data_paths * self.cfg.get('epochs', 100)
        random.seed(233)
        random.shuffle(data_paths)
        rank, world_size = get_rank(), get_world_size()
        per_rank = max(len(data_paths) // world_size, 1)
        data_paths = data_paths[rank * per_rank:(rank + 1) * per_rank]

        self.shared_step_data = fake_step_data(
            share_memory=True, batch_size=self.trajectory_length * self.batch_size)
        self.worker_queue = mp.Queue()
        num_workers = self.cfg.num_workers
        pipes = [mp.Pipe() for _ in range(num_workers)]
        self.pipes_p = [p[0] for p in pipes]
        per_worker = max(len(data_paths) // num_workers, 1)
        self._procs = []
        for i in range(num_workers):
            worker_paths = data_paths[i * per_worker:(i + 1) * per_worker] or data_paths
            proc = mp.Process(target=worker_loop,
                              args=(cfg, worker_paths, pipes[i][1],
                                    self.shared_step_data, self.worker_queue, i),
                              daemon=True)
            proc.start()
            self._procs.append(proc)
        self.worker_indices = [None] * self.batch_size
        for lane in range(self.batch_size):
            self.worker_indices[lane] = self.worker_queue.get()
            self.pipes_p[self.worker_indices[lane]].send(lane)

    def __iter__(self):
        return self

    def __next__(self):
        new_episodes, traj_lens = [], []
        for lane in range(self.batch_size):
            new_episode, traj_len, end_episode = \
                self.pipes_p[self.worker_indices[lane]].recv()
            new_episodes.append(new_episode)
            traj_lens.append(traj_len)
            if end_episode:
                self.worker_indices[lane] = self.worker_queue.get()
        batch = self.shared_step_data
        if self.use_cuda:
            from ..utils.data import to_device
            batch = to_device(batch, self.device)
        for lane in range(self.batch_size):
            self.pipes_p[self.worker_indices[lane]].send(lane)
        batch = dict(batch)
        batch['traj_lens'] = traj_lens
        batch['new_episodes'] = new_episodes
        return batch

    def close(self):
        for p in self._procs:
            p.terminate()
