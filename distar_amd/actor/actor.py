"""Rollout worker (reference `distar/actor/actor.py:23-388`):

  - asks the league for a job (player ids, pipelines, checkpoints, teacher,
    Z), loads models, spawns `env_num` environment processes,
  - per-episode loop: agents step(obs) -> env.step(actions) -> collect_data
    -> ActorComm.send_data; result posted at episode end,
  - model refresh every `actor_model_update_interval` seconds, job rotation
    after `actor_ask_for_job_interval` +/- 30%,
  - crash-tolerant: episode exceptions close the env and continue (actors
    are stateless cattle).

Environment selection: 'mock' (synthetic spec-identical episodes; default in
this offline image) or 'sc2' (real game through envs/env.py, gated on
s2clientprotocol).
"""
import os
import random
import time
import traceback

import torch

from .agent import Agent
from .comm import ActorComm
from ..envs.mock_env import MockSC2Env
from ..utils.checkpoint import CheckpointHelper
from ..utils.config import Config, deep_merge_dicts
from ..utils.log import TextLogger, VariableRecord

DEFAULT_ACTOR_CFG = Config({
    'actor': {
        'env_num': 1, 'episode_num': 2, 'job_type': 'train', 'traj_len': 16,
        'use_cuda': False, 'env_type': 'mock',
        'actor_model_update_interval': 10,
        'actor_ask_for_job_interval': 3600,
        'fake_model': False,
    },
    'env': {'player_num': 2, 'max_episode_steps': 64},
    'common': {'experiment_name': 'actor_default', 'type': 'train'},
})


class Actor:
    def __init__(self, cfg):
        self._whole_cfg = deep_merge_dicts(DEFAULT_ACTOR_CFG, cfg)
        self._cfg = self._whole_cfg.actor
        self._job_type = self._cfg.job_type
        self._comm = ActorComm(self._whole_cfg) \
            if self._whole_cfg.get('communication') else None
        self._logger = TextLogger(
            f'experiments/{self._whole_cfg.common.experiment_name}/log', 'actor')
        self._record = VariableRecord()
        for var in ('agent_time', 'env_time', 'collect_time', 'episode_steps'):
            self._record.register_var(var)
        self._ckpt_helper = CheckpointHelper()
        self._last_model_update = 0
        self._end = False
        self.episodes_done = 0
        self.results = []

    # ------------------------------------------------------------------ job
    def _setup_job(self):
        if self._comm is not None:
            job = self._comm.ask_for_job(
                self._job_type, player_id=self._cfg.get('job_player_id'))
        else:
            # standalone jobs (play/eval or comm-less training) take model
            # checkpoints from actor.model{0,1}_path (bin/play.py contract)
            job = {'player_ids': ['MP0', 'MP1'],
                   'pipelines': ['default', 'default'],
                   'checkpoint_paths': [self._cfg.get('model0_path', 'none'),
                                        self._cfg.get('model1_path', 'none')],
                   'teacher_checkpoint_paths': ['none', 'none'],
                   'z_path': ['3map.json', '3map.json'],
                   'send_data_players': ['MP0'],
                   'update_players': ['MP0'],
                   'env_info': {'map_name': 'KingsCove'}}
        self._job = job
        self._agents = []
        pipelines = job.get('pipelines', ['default'] * len(job['player_ids']))
        for i, player_id in enumerate(job['player_ids']):
            pipeline = pipelines[i] if i < len(pipelines) else 'default'
            if pipeline in ('default', 'bot'):
                agent = Agent(self._whole_cfg, env_id=0)
            else:
                from ..utils.import_helper import import_pipeline_agent
                agent = import_pipeline_agent(pipeline)(self._whole_cfg, env_id=0)
            agent.player_id = player_id
            ckpt = job['checkpoint_paths'][i] if i < len(job['checkpoint_paths']) else 'none'
            if ckpt not in ('none', None) and not self._cfg.fake_model and \
                    getattr(agent, 'HAS_MODEL', False):
                try:
                    raw = torch.load(str(ckpt), map_location='cpu',
                                     weights_only=False)
                    if isinstance(raw, dict) and 'map_name' in raw:
                        # play checkpoints carry their strategy context
                        # (reference actor.py:65-73)
                        job.setdefault('env_info', {})['map_name'] = raw['map_name']
                        agent._cfg = dict(agent._cfg)
                        if 'z_path' in raw:
                            agent._cfg['z_path'] = raw['z_path']
                        if 'fake_reward_prob' in raw:
                            agent._cfg['fake_reward_prob'] = raw['fake_reward_prob']
                        agent.z_idx = raw.get('z_idx')
                    self._ckpt_helper.load(raw, agent.model, strict=False,
                                           need_torch_load=False,
                                           logger_prints=self._logger.info)
                except FileNotFoundError:
                    self._logger.info(f'checkpoint missing: {ckpt}, random init')
            teacher = job.get('teacher_checkpoint_paths', [])
            tpath = teacher[i] if i < len(teacher) else 'none'
            if tpath not in ('none', None) and getattr(agent, 'HAS_TEACHER', False) \
                    and not self._cfg.fake_model and os.path.isfile(str(tpath)):
                from ..models.alphastar.model import Model as _Model
                agent.teacher_model = _Model(self._whole_cfg)
                agent.teacher_model.eval()
                self._ckpt_helper.load(tpath, agent.teacher_model, strict=False,
                                       logger_prints=self._logger.info)
            # DAPO: successive (lagged-self) model per main player (reference
            # actor_comm.py:127-147 reconstructs the league's saved path)
            spaths = job.get('successive_model_paths', [])
            spath = spaths[i] if i < len(spaths) else 'none'
            if self._whole_cfg.get('learner', {}).get('use_dapo', False) and \
                    spath not in ('none', None) and \
                    getattr(agent, 'HAS_MODEL', False) and \
                    not self._cfg.fake_model and os.path.isfile(str(spath)):
                from ..models.alphastar.model import Model as _Model
                agent.successive_model = _Model(self._whole_cfg)
                agent.successive_model.eval()
                self._ckpt_helper.load(spath, agent.successive_model,
                                       strict=False,
                                       logger_prints=self._logger.info)
            # direct (non-batch-server) inference with use_cuda: the agent
            # moves its inputs to cuda, so the models must live there too
            # (batch-inference mode moves them in _start_batch_inference)
            if self._cfg.use_cuda and torch.cuda.is_available() and \
                    not self._cfg.get('gpu_batch_inference', False):
                for m in ('model', 'teacher_model', 'successive_model'):
                    mod = getattr(agent, m, None)
                    if mod is not None:
                        mod.to('cuda')
            self._agents.append(agent)
        return job

    def _make_env(self):
        if self._cfg.env_type == 'sc2':
            from ..envs.env import SC2Env
            return SC2Env(self._whole_cfg)
        return MockSC2Env(self._whole_cfg)

    def _start_batch_inference(self, agent_groups=None):
        """Shared-slab batched inference (reference actor.py:268-299): ONE
        server per PLAYER (each serving that player's model — and its teacher
        slab on train jobs — across all env workers' slots); agents switch to
        writing their obs into their server's slab and polling the signal."""
        import threading
        from .batch_inference import BatchInferenceServer
        device = 'cuda' if (self._cfg.use_cuda and torch.cuda.is_available()) \
            else 'cpu'
        agent_groups = agent_groups or [self._agents]
        env_n = len(agent_groups)
        self._batch_servers, self._batch_threads = [], []
        for i, agent0 in enumerate(self._agents):
            if not getattr(agent0, 'HAS_MODEL', False):
                self._batch_servers.append(None)
                continue
            teacher = agent0.teacher_model \
                if 'train' in self._job_type and agent0.teacher_model is not None \
                else None
            server = BatchInferenceServer(
                agent0.model.to(device), env_num=env_n, device=device,
                teacher_model=teacher.to(device) if teacher is not None else None)
            for env_id, group in enumerate(agent_groups):
                group[i].attach_batch_inference(server, env_id)
            t = threading.Thread(target=server.run, daemon=True)
            t.start()
            self._batch_servers.append(server)
            self._batch_threads.append(t)
        self._batch_server = next((x for x in self._batch_servers if x), None)
        return self._batch_servers

    def _stop_batch_inference(self):
        for srv in getattr(self, '_batch_servers', []):
            if srv is not None:
                srv.stop()
        for t in getattr(self, '_batch_threads', []):
            t.join(timeout=5)

    def _update_models(self):
        if self._comm is None:
            return
        now = time.time()
        if now - self._last_model_update < self._cfg.actor_model_update_interval:
            return
        self._last_model_update = now
        for agent in self._agents:
            if agent.player_id in self._job.get('update_players', []):
                payload = self._comm.pull_model(agent.player_id, timeout=2)
                if payload:
                    agent.model.load_state_dict(payload['model'], strict=False)
                    agent.set_model_last_iter(payload.get('model_last_iter', 0))

    # -------------------------------------------------------------- episode
    def _run_episode(self, env, agents=None):
        agents = agents if agents is not None else self._agents
        obs = env.reset()
        for i, agent in enumerate(agents):
            agent.reset(map_name=self._job['env_info'].get('map_name', 'KingsCove'),
                        race='zerg', opponent_race='zerg', obs=obs.get(i))
        done = False
        episode_steps = 0
        last_obs = obs
        while not done and not self._end:
            t0 = time.time()
            actions = {}
            for i, agent in enumerate(agents):
                if i in last_obs:
                    actions[i] = agent.step(last_obs[i])[0]
            t1 = time.time()
            obs, rewards, done, infos = env.step(actions)
            t2 = time.time()
            for i, agent in enumerate(agents):
                if i not in last_obs:
                    continue
                traj = agent.collect_data(obs.get(i), rewards.get(i, 0), done, i)
                if traj is not None and self._comm is not None and \
                        agent.player_id in self._job.get('send_data_players', []):
                    self._comm.send_data(traj, agent.player_id)
            self._record.update_var({'agent_time': t1 - t0,
                                     'env_time': t2 - t1,
                                     'collect_time': time.time() - t2})
            episode_steps += 1
            last_obs = {**last_obs, **obs}
            self._update_models()
        self._record.update_var({'episode_steps': episode_steps})
        result = self._build_result(rewards, agents)
        if self._comm is not None:
            self._comm.send_result(result)
        self.results.append(result)
        return result

    def _build_result(self, rewards, agents=None):
        agents = agents if agents is not None else self._agents
        result = {'game_steps': agents[0]._game_step,
                  'game_iters': agents[0]._iter_count,
                  'game_duration': 0}
        ids = self._job['player_ids']
        for i, agent in enumerate(agents):
            opp = ids[1 - i] if len(ids) > 1 else ids[0]
            side = {'player_id': agent.player_id, 'opponent_id': opp,
                    'winloss': float(rewards.get(i, 0))}
            side.update(agent.get_stat_data())
            if hasattr(agent, 'get_unit_num_info'):
                side.update(agent.get_unit_num_info())
            result[str(i)] = side
        return result

    # ------------------------------------------------------------------ run
    def _env_loop(self, agents, episode_num):
        """One environment worker: episodes until quota/stop (the reference
        forks `env_num` processes, `actor.py:301-319`; here each worker is a
        thread sharing the job's model weights — inference is no-grad and
        per-agent recurrent state lives on the Agent)."""
        env = self._make_env()
        crashes = 0
        while not self._end and (episode_num < 0 or self.episodes_done < episode_num):
            try:
                self._run_episode(env, agents)
                self.episodes_done += 1
                crashes = 0
            except Exception:  # noqa: BLE001 - actors are cattle
                traceback.print_exc()
                crashes += 1
                if crashes >= 20:   # systematic failure, not a flaky episode
                    self._logger.info('20 consecutive episode crashes, stopping worker')
                    break
                try:
                    env.close()
                except Exception:  # noqa: BLE001
                    pass
                env = self._make_env()
        env.close()

    def _clone_agents(self, env_id):
        """Per-env agent wrappers sharing the job's model objects."""
        clones = []
        for agent in self._agents:
            clone = type(agent)(self._whole_cfg, env_id=env_id)
            clone.player_id = agent.player_id
            if getattr(agent, 'HAS_MODEL', False):
                clone.model = agent.model
                clone.teacher_model = agent.teacher_model
                clone.successive_model = agent.successive_model
            clones.append(clone)
        return clones

    def run(self):
        import threading
        self._setup_job()
        env_num = self._cfg.env_num
        episode_num = self._cfg.episode_num
        agent_groups = [self._agents] + \
            [self._clone_agents(i) for i in range(1, env_num)]
        if self._cfg.get('gpu_batch_inference', False):
            self._start_batch_inference(agent_groups)
        job_deadline = time.time() + self._cfg.actor_ask_for_job_interval * \
            (1 + 0.3 * (2 * random.random() - 1))
        if env_num <= 1:
            env = self._make_env()
            crashes = 0
            while not self._end and \
                    (episode_num < 0 or self.episodes_done < episode_num):
                try:
                    self._run_episode(env)
                    self.episodes_done += 1
                    crashes = 0
                except Exception:  # noqa: BLE001 - actors are cattle
                    traceback.print_exc()
                    crashes += 1
                    if crashes >= 20:
                        self._logger.info('20 consecutive episode crashes, stopping')
                        break
                    try:
                        env.close()
                    except Exception:  # noqa: BLE001
                        pass
                    env = self._make_env()
                if time.time() > job_deadline:
                    self._setup_job()
                    if self._cfg.get('gpu_batch_inference', False):
                        # new job -> new agents/models: restart the slab servers
                        self._stop_batch_inference()
                        self._start_batch_inference()
                    job_deadline = time.time() + self._cfg.actor_ask_for_job_interval
            env.close()
            self._stop_batch_inference()
            return self.results
        workers = []
        for env_id in range(env_num):
            t = threading.Thread(target=self._env_loop,
                                 args=(agent_groups[env_id], episode_num),
                                 daemon=True)
            t.start()
            workers.append(t)
        while any(t.is_alive() for t in workers):
            for t in workers:
                t.join(timeout=0.5)
            self._update_models()
        self._stop_batch_inference()
        return self.results

    def close(self):
        self._end = True
        self._stop_batch_inference()
