"""SC2 multi-agent environment (reference `distar/envs/env.py:96-449`).

`SC2Env` keeps the reference's public contract — ``reset() -> {agent: obs}``,
``step(actions)`` with per-agent skip-step scheduling and 0-3 random delay
steps, win/loss outcome extraction, replay saving, and a full game restart
every 10 episodes — implemented against a `RemoteController` protocol
abstraction (envs/protocol.py).

The StarCraft II protobuf bindings (s2clientprotocol) and game binary are
not shipped in this image; constructing `SC2Env` without them raises
ImportError with instructions, and `MockSC2Env` (mock_env.py) provides the
spec-identical no-binary environment used by tests and synthetic rollouts
(the same split the reference makes with `pysc2/env/mock_sc2_env.py`).
"""
import random
import time

from .map_info import get_map_size
from .protocol import SC2_PROTO_AVAILABLE, RemoteController, launch_game_process

RESTART_EPISODE_INTERVAL = 10
DELAY_WEIGHTS = [1, 1, 1, 1]           # 0-3 extra latency steps


class SC2Env:
    def __init__(self, cfg, seed=None):
        if not SC2_PROTO_AVAILABLE:
            raise ImportError(
                'SC2Env needs the s2clientprotocol protobuf bindings and a '
                'StarCraft II install; neither ships in this offline image. '
                'Use distar_amd.envs.MockSC2Env for spec-identical synthetic '
                'episodes, or install s2clientprotocol + SC2 and point '
                'SC2PATH at the install.')
        self._whole_cfg = cfg
        self._cfg = cfg.env
        self._agent_num = self._cfg.get('player_num', 2)
        self._map_name = self._cfg.get('map_name', 'KingsCove')
        self.map_size = get_map_size(self._map_name)
        self._rng = random.Random(seed)
        self._save_replay_episodes = self._cfg.get('save_replay_episodes', 0)
        self._replay_dir = self._cfg.get('replay_dir', '.')
        self._game_procs = []
        self._controllers = []
        self._episode_count = 0
        self._episode_steps = 0
        self._next_obs_step = [0] * self._agent_num
        self._launched = False

    # ------------------------------------------------------------ lifecycle
    def _launch_game(self):
        self.close()
        ports = []
        for _ in range(self._agent_num):
            proc, port = launch_game_process(self._whole_cfg)
            self._game_procs.append(proc)
            ports.append(port)
            self._controllers.append(RemoteController('127.0.0.1', port))
        self._create_join(ports)
        self._launched = True

    def _create_join(self, ports):
        self._controllers[0].create_game(self._map_name, self._agent_num, ports)
        for i, ctrl in enumerate(self._controllers):
            ctrl.join_game(race=self._cfg.get('races', ['zerg'] * self._agent_num)[i],
                           ports=ports)

    def reset(self):
        for attempt in range(10):
            try:
                if not self._launched or \
                        self._episode_count % RESTART_EPISODE_INTERVAL == 0:
                    self._launch_game()
                else:
                    for ctrl in self._controllers:
                        ctrl.restart_game()
                break
            except (ConnectionError, OSError) as e:
                print(f'[SC2Env] launch failed ({e!r}), retry {attempt + 1}/10')
                time.sleep(1)
        else:
            raise ConnectionError('SC2 launch failed after 10 retries')
        self._episode_count += 1
        self._episode_steps = 0
        self._next_obs_step = [0] * self._agent_num
        return {i: ctrl.observe() for i, ctrl in enumerate(self._controllers)}

    # ----------------------------------------------------------------- step
    def step(self, actions):
        """Per-agent skip scheduling (reference `envs/env.py:333-375`): each
        action carries skip_steps; the env advances to min(next_obs_step)
        plus 0-3 random delay steps, then observes agents that are due."""
        for idx, action in (actions or {}).items():
            if action is None:
                continue
            self._controllers[idx].acts(action)
            self._next_obs_step[idx] = self._episode_steps + \
                int(action.get('skip_steps', 0)) + 1
        target = min(self._next_obs_step)
        delay = self._rng.choices([0, 1, 2, 3], weights=DELAY_WEIGHTS)[0]
        step_count = max(target + delay - self._episode_steps, 1)
        for ctrl in self._controllers:
            ctrl.step(step_count)
        self._episode_steps += step_count
        obs, rewards, done = {}, {}, False
        for i, ctrl in enumerate(self._controllers):
            if self._next_obs_step[i] <= self._episode_steps:
                o = ctrl.observe()
                obs[i] = o
                outcome = ctrl.outcome(o)
                if outcome is not None:
                    done = True
                    rewards[i] = outcome
        if done:
            for i in obs:
                rewards.setdefault(i, -sum(rewards.values()))
            if self._save_replay_episodes and \
                    self._episode_count % self._save_replay_episodes == 0:
                self.save_replay()
        else:
            rewards = {i: 0 for i in obs}
        infos = {i: {'episode_steps': self._episode_steps} for i in obs}
        return obs, rewards, done, infos

    def save_replay(self):
        self._controllers[0].save_replay(self._replay_dir)

    def close(self):
        for ctrl in self._controllers:
            try:
                ctrl.quit()
            except Exception:  # noqa: BLE001
                pass
        for proc in self._game_procs:
            try:
                proc.kill()
            except Exception:  # noqa: BLE001
                pass
        self._controllers = []
        self._game_procs = []
        self._launched = False
