"""Actor/learner communication.

Functional parity with the reference's
`ctools/worker/actor/actor_comm.py:47-258` (job ask/result over the league
HTTP API, model refresh via Adapter pull, trajectory push with backlog
warning) and `ctools/worker/learner/learner_comm.py:20-171` (learner
registration, policy-only model publishing, train-info posts with
league-ordered checkpoint reset + broadcast).
"""
import os
import time


from ..data.adapter import Adapter
from ..parallel.dist import broadcast, get_rank, get_world_size, is_initialized
from ..utils.http import post_json


class ActorComm:
    def __init__(self, cfg, adapter=None):
        self._whole_cfg = cfg
        comm = cfg.get('communication', {})
        self._league_url = 'http://{}:{}'.format(
            comm.get('league_ip', '127.0.0.1'), comm.get('league_port', 0))
        self.adapter = adapter or Adapter(cfg=cfg)
        self._traj_compress = comm.get('traj_compress', False)
        self.job = {}

    def ask_for_job(self, job_type='train', player_id=None):
        req = {'job_type': job_type}
        if player_id:
            req['player_id'] = player_id
        self.job = post_json(self._league_url + '/league/actor_ask_for_job',
                             req, retries=10, backoff=1.0)
        return self.job

    def send_result(self, result_info):
        return post_json(self._league_url + '/league/actor_send_result',
                         result_info)

    def pull_model(self, player_id, timeout=10):
        """Fetch the freshest published policy state_dict for a player."""
        out = self.adapter.pull(token=player_id + 'model', fs_type='pickle',
                                sleep_time=0.2, size=1, timeout=timeout)
        return out[0] if out else None

    def send_data(self, traj_data, player_id):
        """Push one trajectory; warn on backlog (reference
        actor_comm.py:218-225)."""
        token = player_id + 'traj'
        if self.adapter.full(token):
            print(f'[ActorComm] trajectory backlog full for {token}')
        # zlib-1 costs ~5-10x nppickle dump time for 1.63x size (see
        # profiles/r01_serialize.md) -> uncompressed by default; turn on for
        # bandwidth-constrained actor fleets via communication.traj_compress
        self.adapter.push(traj_data, token=token, fs_type='nppickle',
                          compress=self._traj_compress)


class LearnerComm:
    def __init__(self, cfg, adapter=None):
        self._whole_cfg = cfg
        comm = cfg.get('communication', {})
        self._league_url = 'http://{}:{}'.format(
            comm.get('league_ip', '127.0.0.1'), comm.get('league_port', 0))
        self.adapter = adapter or Adapter(cfg=cfg)
        self.player_id = cfg.get('learner', {}).get('player_id', 'MP0')
        self._traj_compress = comm.get('traj_compress', False)
        self._send_model_count = 0

    def register_learner(self, learner):
        resp = post_json(self._league_url + '/league/register_learner',
                         {'player_id': self.player_id,
                          'ip': '127.0.0.1', 'port': 0,
                          'rank': learner.rank, 'world_size': learner.world_size},
                         retries=10, backoff=1.0)
        ckpt = resp.get('ckpt_path')
        if ckpt and ckpt != 'none' and os.path.isfile(ckpt):
            learner._load_path = ckpt
        return resp

    @staticmethod
    def strip_value_keys(state_dict):
        """Actors run policy-only models (reference learner_comm.py:74)."""
        return {k: v for k, v in state_dict.items()
                if not k.startswith(('value_networks', 'value_encoder'))}

    def send_model(self, learner):
        model = getattr(learner.model, 'module', learner.model)
        state_dict = {k: v.detach().cpu() for k, v in
                      self.strip_value_keys(model.state_dict()).items()}
        payload = {'model': state_dict, 'model_last_iter': learner.last_iter.val}
        self.adapter.push(payload, token=self.player_id + 'model',
                          fs_type='pickle', compress=self._traj_compress)
        self._send_model_count += 1

    def send_train_info(self, learner):
        frames = int(learner.whole_cfg.learner.data.batch_size *
                     learner.whole_cfg.learner.data.trajectory_length *
                     get_world_size())
        ckpt_path = learner.save_checkpoint(self.player_id) or 'none'
        resp = post_json(self._league_url + '/league/learner_send_train_info',
                         {'player_id': self.player_id, 'train_steps': frames,
                          'checkpoint_path': ckpt_path})
        reset_path = resp.get('reset_checkpoint_path', 'none')
        if reset_path != 'none' and os.path.isfile(reset_path):
            # league-ordered reset: rank 0 loads, then broadcasts params
            model = getattr(learner.model, 'module', learner.model)
            learner.checkpoint_helper.load(reset_path, model, strict=False,
                                           logger_prints=learner.info)
            if is_initialized() and get_world_size() > 1:
                for p in model.parameters():
                    broadcast(p.data, src=0)
            learner.info(f'league reset -> {reset_path}')
        return resp
