"""Synthetic dataloaders for benchmarks and `job_type != 'train'` runs.

Counterpart of the reference's `sl_dataloader.FakeDataloader` and the RL
fake-data path (`rl_learner.py:192-197`), built on our fake-batch builders.
A small pool of pre-built batches is cycled (optionally pre-staged on the
GPU) so the learner benchmark measures the learner, not the Python cost of
synthesizing observations.
"""
import itertools


from ..lib.fake_data import fake_rl_learner_data, fake_sl_batch
from ..utils.data import to_device


class FakeSLDataloader:
    def __init__(self, cfg, pool_size=2, device=None):
        data_cfg = cfg.learner.data
        self.batch_size = data_cfg.batch_size
        self.traj_len = data_cfg.trajectory_length
        self.device = device
        pool = [fake_sl_batch(self.batch_size, self.traj_len, seed=i)
                for i in range(pool_size)]
        if device is not None and device != 'cpu':
            pool = [to_device(b, device) for b in pool]
        self._cycle = itertools.cycle(pool)

    def __iter__(self):
        return self

    def __next__(self):
        batch = next(self._cycle)
        # fresh lists: the learner pops new_episodes each iteration
        out = dict(batch)
        out['new_episodes'] = [False] * self.batch_size
        out['traj_lens'] = [self.traj_len] * self.batch_size
        return out


class FakeRLDataloader:
    def __init__(self, cfg, pool_size=2, device=None):
        data_cfg = cfg.learner.data
        self.batch_size = data_cfg.batch_size
        self.unroll_len = data_cfg.trajectory_length
        entity_range = tuple(data_cfg.get('fake_entity_range', (64, 256)))
        self.device = device
        pool = [fake_rl_learner_data(self.batch_size, self.unroll_len,
                                     entity_num_range=entity_range, seed=i)
                for i in range(pool_size)]
        if device is not None and device != 'cpu':
            pool = [to_device(b, device) for b in pool]
        self._cycle = itertools.cycle(pool)

    def __iter__(self):
        return self

    def __next__(self):
        return dict(next(self._cycle))

    def close(self):
        pass

    def reset_comm(self):
        pass
