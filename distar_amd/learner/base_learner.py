"""Training-loop skeleton.

Functional parity with the reference's
`ctools/worker/learner/base_learner.py:24-383`: hooks(before_run) ->
loop{fetch -> before_iter -> _train -> after_iter} -> after_run, the whole
run wrapped in @auto_checkpoint; distributed init + DistModule wrap; timers
around data-fetch and train; Adam + MultiStepLR with warmup.

MI355X specifics: `DistModule` is the bucketed backward-overlapped RCCL
allreduce (parallel/ddp.py); timers are hipEvent-backed on device; the model
runs with bf16 autocast when `learner.use_amp` (default on GPU).
"""
import os
import time

import torch

from .hooks import build_learner_hook_by_cfg, add_learner_hook
from ..parallel.ddp import DistModule
from ..parallel.dist import dist_init, get_rank, get_world_size
from ..utils.checkpoint import CheckpointHelper, CountVar, auto_checkpoint
from ..utils.config import Config, deep_merge_dicts, read_config
from ..utils.log import build_logger
from ..utils.timing import EasyTimer

DEFAULT_LEARNER_CFG = Config({
    'learner': {
        'load_path': '', 'use_cuda': None, 'use_distributed': False,
        'use_amp': None,
        'learning_rate': 1e-3, 'weight_decay': 0.0,
        'lr_decay_milestones': [], 'lr_decay': 1.0, 'warmup_iters': 0,
        'grad_clip': {'type': 'pytorch_norm', 'threshold': 1.0},
        'job_type': 'train',
        'hook': {
            'before_run': {'load_ckpt': {}},
            'after_iter': {
                'log_reduce': {'priority': 10},
                'log_show': {'priority': 30, 'ext_args': {'freq': 100}},
                'save_ckpt': {'ext_args': {'freq': 1000}},
            },
            'after_run': {},
        },
    },
    'common': {'experiment_dir': 'experiments', 'experiment_name': 'default'},
})


class BaseLearner:
    _name = 'BaseLearner'

    def __init__(self, cfg, method=None, init_method=None, rank=0, world_size=1):
        self._whole_cfg = deep_merge_dicts(DEFAULT_LEARNER_CFG, cfg)
        learner_cfg = self._whole_cfg.learner
        self._use_distributed = learner_cfg.use_distributed
        if self._use_distributed:
            dist_init(method=method or 'torch', rank=rank, world_size=world_size,
                      init_method=init_method)
        self._rank = get_rank()
        self._world_size = get_world_size()
        use_cuda = learner_cfg.use_cuda
        self._use_cuda = torch.cuda.is_available() if use_cuda is None else use_cuda
        self._device = torch.cuda.current_device() if self._use_cuda else 'cpu'
        use_amp = learner_cfg.use_amp
        self._use_amp = self._use_cuda if use_amp is None else use_amp
        self._load_path = learner_cfg.load_path
        self._experiment_name = self._whole_cfg.common.experiment_name
        self._exp_dir = os.path.join(self._whole_cfg.common.experiment_dir,
                                     self._experiment_name)

        self._timer = EasyTimer(cuda=self._use_cuda)
        self._data_timer = EasyTimer(cuda=self._use_cuda)
        self._last_iter = CountVar(0)
        self._log_buffer = {}
        self._checkpoint_helper = CheckpointHelper(self._rank)
        self._logger, self._scalar_logger, self._record = build_logger(
            self._whole_cfg, name=self._name, rank=self._rank)
        if self._rank == 0:
            # back up the fully-merged config for reproducibility (reference
            # bin/rl_train.py:27-42 copies configs into the experiment dir)
            from ..utils.config import save_config
            cfg_dir = os.path.join(self._exp_dir, 'config_backup')
            os.makedirs(cfg_dir, exist_ok=True)
            try:
                save_config(self._whole_cfg,
                            os.path.join(cfg_dir, f'{self._name}_whole_config.yaml'))
            except Exception as e:  # noqa: BLE001 - never block training on this
                self.info(f'config backup failed: {e!r}')
        self._setup_model()
        if self._use_cuda:
            self._model = self._model.cuda()
        if self._use_distributed:
            self._model = DistModule(self._model,
                                     bucket_cap_mb=learner_cfg.get('bucket_cap_mb', 64))
        self._setup_optimizer()
        self._setup_loss()
        self._setup_dataloader()
        self._hooks = build_learner_hook_by_cfg(learner_cfg.hook)
        self._setup_extra_hooks()
        self._register_stats()
        self._end_flag = False

    # --------------------------------------------------------- overridables
    def _setup_model(self):
        raise NotImplementedError

    def _setup_loss(self):
        raise NotImplementedError

    def _setup_dataloader(self):
        raise NotImplementedError

    def _setup_extra_hooks(self):
        pass

    def _register_stats(self):
        for var in ('cur_lr', 'data_time', 'train_time', 'forward_time',
                    'backward_time', 'total_loss', 'gradient'):
            self._record.register_var(var)
            if self._scalar_logger is not None:
                self._scalar_logger.register_var(var)
        if hasattr(self, '_loss') and hasattr(self._loss, 'register_stats') \
                and self._scalar_logger is not None:
            self._loss.register_stats(self._record, self._scalar_logger)

    def _setup_optimizer(self):
        cfg = self._whole_cfg.learner
        self._optimizer = torch.optim.Adam(
            self.model.parameters(), lr=cfg.learning_rate,
            weight_decay=cfg.weight_decay, fused=self._use_cuda or None)
        milestones = list(cfg.lr_decay_milestones)
        if not milestones and cfg.get('lr_decay_interval'):
            # reference-config compatibility (base_learner.py:171-176): decay
            # by lr_decay every lr_decay_interval iterations, 40 steps out
            interval = int(cfg.lr_decay_interval)
            milestones = list(range(interval, interval * 40, interval))
        decay = torch.optim.lr_scheduler.MultiStepLR(
            self._optimizer, milestones=milestones, gamma=cfg.lr_decay)
        warmup_iters = int(cfg.get('warm_up_steps', cfg.warmup_iters))
        if warmup_iters > 0:
            warmup = torch.optim.lr_scheduler.LinearLR(
                self._optimizer, start_factor=1e-3, total_iters=warmup_iters)
            self._lr_scheduler = torch.optim.lr_scheduler.SequentialLR(
                self._optimizer, [warmup, decay], milestones=[warmup_iters])
        else:
            self._lr_scheduler = decay

    def _train(self, data):
        raise NotImplementedError

    # ----------------------------------------------------------------- api
    def call_hook(self, position):
        for hook in self._hooks[position]:
            hook(self)

    def register_hook(self, hook):
        add_learner_hook(self._hooks, hook)

    @auto_checkpoint
    def run(self, max_iterations=None):
        max_iterations = max_iterations or \
            self._whole_cfg.learner.get('max_iterations', int(1e9))
        self.call_hook('before_run')
        while self._last_iter.val < max_iterations and not self._end_flag:
            with self._data_timer:
                data = next(self._dataloader)
            self._log_buffer['data_time'] = self._data_timer.value
            self.call_hook('before_iter')
            with self._timer:
                self._train(data)
            self._log_buffer['train_time'] = self._timer.value
            self._last_iter.add(1)
            self.call_hook('after_iter')
        self.call_hook('after_run')

    def close(self):
        self._end_flag = True

    def save_checkpoint(self, player_id=None):
        if self._rank != 0:
            return None
        sub = f'{player_id}/' if player_id else ''
        ckpt_dir = os.path.join(self._exp_dir, sub + 'checkpoint')
        name = f'{self._experiment_name}_' + \
            (f'{player_id}_' if player_id else '') + \
            f'iteration_{self._last_iter.val}.pth.tar'
        path = os.path.join(ckpt_dir, name)
        model = self._model.module if isinstance(self._model, DistModule) else self._model
        self._checkpoint_helper.save(path, model, optimizer=self._optimizer,
                                     last_iter=self._last_iter)
        self.info(f'saved checkpoint {path}')
        return path

    def info(self, msg):
        if self._logger is not None:
            self._logger.info(msg)

    # ------------------------------------------------------------- accessors
    @property
    def model(self):
        return self._model

    @property
    def optimizer(self):
        return self._optimizer

    @property
    def lr_scheduler(self):
        return self._lr_scheduler

    @property
    def last_iter(self):
        return self._last_iter

    @property
    def log_buffer(self):
        return self._log_buffer

    @property
    def record(self):
        return self._record

    @property
    def scalar_logger(self):
        return self._scalar_logger

    @property
    def rank(self):
        return self._rank

    @property
    def world_size(self):
        return self._world_size

    @property
    def load_path(self):
        return self._load_path

    @property
    def checkpoint_helper(self):
        return self._checkpoint_helper

    @property
    def whole_cfg(self):
        return self._whole_cfg
