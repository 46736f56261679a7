"""SC2 multi-agent environment (reference `distar/envs/env.py:96-449`).

`SC2Env` keeps the reference's public contract — ``reset() -> {agent: obs}``,
``step(actions)`` with per-agent skip-step scheduling and 0-3 random delay
steps, win/loss outcome extraction, replay saving, and a full game restart
every 10 episodes — on top of the repo-native protocol stack
(envs/protocol.py): one SC2 process + RemoteController per agent, game/base
port sets reserved via portspicker (NOT the websocket listen ports —
reference `envs/env.py:211-274`), Computer-player (bot) opponents with
difficulty/ai_build, and realtime mode for play.

Requires a StarCraft II install (SC2PATH); the protobuf wire layer is
self-contained (lib/sc2_protos.py).  `MockSC2Env` (mock_env.py) remains the
spec-identical no-binary environment used by tests and synthetic rollouts.
"""
import random
import time

from .map_info import get_map_size
from .portspicker import pick_unused_ports, return_ports
from .protocol import RemoteController, launch_game_process

RESTART_EPISODE_INTERVAL = 10
DELAY_WEIGHTS = [1, 1, 1, 1]           # 0-3 extra latency steps


class SC2Env:
    def __init__(self, cfg, seed=None):
        self._whole_cfg = cfg
        self._cfg = cfg.env
        self._agent_num = self._cfg.get('player_num', 2)
        self._bots = list(self._cfg.get('bots', []))
        # bot_difficulty shorthand: one built-in bot opponent
        if not self._bots and self._cfg.get('bot_difficulty'):
            self._bots = [{'race': self._cfg.get('bot_race', 'zerg'),
                           'difficulty': self._cfg.get('bot_difficulty')}]
        if self._bots:                   # bots replace absent agents
            self._agent_num = max(1, self._agent_num - len(self._bots))
        self._realtime = bool(self._cfg.get('realtime', False))
        self._map_name = self._cfg.get('map_name', 'KingsCove')
        self.map_size = get_map_size(self._map_name)
        self._rng = random.Random(seed)
        self._random_seed = seed
        self._save_replay_episodes = self._cfg.get('save_replay_episodes', 0)
        self._replay_dir = self._cfg.get('replay_dir', '.')
        self._game_procs = []
        self._controllers = []
        self._game_ports = []
        self._episode_count = 0
        self._episode_steps = 0
        self._next_obs_step = [0] * self._agent_num
        self._launched = False

    # ------------------------------------------------------------ lifecycle
    def _launch_game(self):
        self.close()
        for _ in range(self._agent_num):
            proc, port = launch_game_process(self._whole_cfg)
            self._game_procs.append(proc)
            self._controllers.append(RemoteController('127.0.0.1', port))
        self._create_join()
        self._launched = True

    def _player_setup(self):
        players = [{'type': 'participant'} for _ in range(self._agent_num)]
        for bot in self._bots:
            players.append({'type': 'computer',
                            'race': bot.get('race', 'random'),
                            'difficulty': bot.get('difficulty', 'very_hard'),
                            'ai_build': bot.get('ai_build')})
        return players

    def _create_join(self):
        races = self._cfg.get('races', ['zerg'] * self._agent_num)
        self._controllers[0].create_game(
            map_name=self._map_name, players=self._player_setup(),
            realtime=self._realtime, random_seed=self._random_seed)
        ports = None
        if self._agent_num > 1:
            # 2 server ports + 2 per additional client, freshly reserved —
            # the SC2 websocket listen ports are already bound and must not
            # be reused here (reference `envs/env.py:258-266`)
            flat = pick_unused_ports(2 * self._agent_num)
            self._game_ports = flat
            ports = {'server': (flat[0], flat[1]),
                     'clients': [(flat[2 + 2 * i], flat[3 + 2 * i])
                                 for i in range(self._agent_num - 1)]}
        for i, ctrl in enumerate(self._controllers):
            ctrl.join_game(race=races[i], ports=ports,
                           minimap_resolution=self.map_size)

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
        self._game_infos = [ctrl.game_info() for ctrl in self._controllers]
        return {i: self._wrap_obs(i, ctrl.observe())
                for i, ctrl in enumerate(self._controllers)}

    def _wrap_obs(self, idx, obs):
        obs['game_info_proto'] = self._game_infos[idx]
        return obs

    # --------------------------------------------------------- action protos
    def transform_action(self, actions):
        """Agent action dicts -> raw-action protos (reference
        `envs/env.py:457-483`): `{'func_id','skip_steps','queued',
        'unit_tags','target_unit_tag','location'}` routed by the raw
        function table's function_type."""
        from .protocol import sc_pb
        from ..lib.features import RAW_FUNC_BY_ID
        sc2_actions = []
        skip = None
        for a in actions:
            skip = int(a.get('skip_steps', 0)) if skip is None \
                else min(skip, int(a.get('skip_steps', 0)))
            f = RAW_FUNC_BY_ID[int(a['func_id'])]
            ftype = f['function_type']
            if ftype == 'raw_no_op':
                continue
            act = sc_pb.Action()
            if ftype == 'raw_move_camera':
                loc = a['location']
                act.action_raw.camera_move.center_world_space.x = float(loc[0])
                act.action_raw.camera_move.center_world_space.y = float(loc[1])
            elif ftype == 'raw_autocast':
                ta = act.action_raw.toggle_autocast
                ta.ability_id = f['ability_id']
                ta.unit_tags.extend(int(t) for t in a.get('unit_tags', []))
            else:
                uc = act.action_raw.unit_command
                uc.ability_id = f['ability_id']
                uc.queue_command = bool(a.get('queued', 0))
                uc.unit_tags.extend(int(t) for t in a.get('unit_tags', []))
                if ftype == 'raw_cmd_pt':
                    loc = a['location']
                    uc.target_world_space_pos.x = float(loc[0])
                    uc.target_world_space_pos.y = float(loc[1])
                elif ftype == 'raw_cmd_unit':
                    uc.target_unit_tag = int(a['target_unit_tag'])
            sc2_actions.append(act)
        return sc2_actions, (skip or 0)

    # ----------------------------------------------------------------- step
    def step(self, actions):
        """Per-agent skip scheduling (reference `envs/env.py:333-375`): each
        action carries skip_steps; the env advances to min(next_obs_step)
        plus 0-3 random delay steps, then observes agents that are due.
        Accepts pre-built `{'raw_actions': [...]}` or agent action dicts
        (single or list) with `func_id`."""
        for idx, action in (actions or {}).items():
            if action is None:
                continue
            if isinstance(action, dict) and 'raw_actions' in action:
                protos = action['raw_actions']
                skip = int(action.get('skip_steps', 0))
            else:
                alist = action if isinstance(action, list) else [action]
                protos, skip = self.transform_action(alist)
            self._controllers[idx].acts({'raw_actions': protos})
            self._next_obs_step[idx] = self._episode_steps + skip + 1
        target = min(self._next_obs_step)
        delay = self._rng.choices([0, 1, 2, 3], weights=DELAY_WEIGHTS)[0]
        step_count = max(target + delay - self._episode_steps, 1)
        if not self._realtime:
            for ctrl in self._controllers:
                ctrl.step(step_count)
        self._episode_steps += step_count
        obs, rewards, done = {}, {}, False
        for i, ctrl in enumerate(self._controllers):
            if self._next_obs_step[i] <= self._episode_steps:
                o = self._wrap_obs(i, ctrl.observe())
                obs[i] = o
                outcome = ctrl.outcome(o)
                if outcome is not None:
                    done = True
                    rewards[i] = outcome
        if done:
            # game over: EVERY agent gets its final observation + outcome,
            # due or not (the skip schedule no longer applies)
            for i, ctrl in enumerate(self._controllers):
                if i not in obs:
                    o = self._wrap_obs(i, ctrl.observe())
                    obs[i] = o
                    outcome = ctrl.outcome(o)
                    if outcome is not None:
                        rewards[i] = outcome
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
        if self._game_ports:
            return_ports(self._game_ports)
            self._game_ports = []
        self._controllers = []
        self._game_procs = []
        self._launched = False
