"""StarCraft II protocol client: versioned binary launch + the full
RemoteController request surface.

Replaces the reference's vendored pysc2 launch/controller layer
(`distar/pysc2/run_configs/platforms.py`,
`distar/pysc2/lib/remote_controller.py:127-350`) with an MI355X-repo-native
stack built on stdlib sockets (utils/ws.py), wire-compatible protobuf
messages (lib/sc2_protos.py — the official `s2clientprotocol` package is
preferred automatically when installed), and the published game-version ->
(base build, data version) table (assets/sc2_versions.json).

Surface: create_game (Participant AND Computer/bot players, realtime),
join_game with portspicker-reserved game/base port sets, observe (with
action-result ints, disable_fog, target_game_loop), act/acts, step,
game_info, data/data_raw, ping, replay_info, start_replay, save_replay,
save_map, available_maps, restart_game, leave, quit — each checking the
per-response error field and tracking game status.
"""
import json
import os
import subprocess
import time

try:
    from ..lib.sc2_protos import get_protos
    sc_pb, PROTO_SOURCE = get_protos()
    SC2_PROTO_AVAILABLE = True
except ImportError:                      # pragma: no cover - protobuf absent
    sc_pb, PROTO_SOURCE = None, None
    SC2_PROTO_AVAILABLE = False

from ..utils.ws import WebSocket, WebSocketError
from .portspicker import pick_unused_ports, return_ports   # noqa: F401

RACES = {'zerg': 2, 'terran': 1, 'protoss': 3, 'random': 4}
DIFFICULTIES = {
    'very_easy': 1, 'easy': 2, 'medium': 3, 'medium_hard': 4, 'hard': 5,
    'harder': 6, 'very_hard': 7, 'cheat_vision': 8, 'cheat_money': 9,
    'cheat_insane': 10,
}
AI_BUILDS = {'random': 1, 'rush': 2, 'timing': 3, 'power': 4, 'macro': 5,
             'air': 6}

_ASSETS = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                       'assets')
_VERSIONS = None


class ProtocolError(ConnectionError):
    """SC2 rejected a request (response carried an error)."""


def known_versions():
    """game-version string -> {'base_build', 'data_version'}
    (Blizzard's published build info; assets/sc2_versions.json)."""
    global _VERSIONS
    if _VERSIONS is None:
        with open(os.path.join(_ASSETS, 'sc2_versions.json')) as f:
            _VERSIONS = json.load(f)
    return _VERSIONS


def version_info(game_version):
    """Resolve '4.10.0' / '4.10.0.75689' / None to a version record
    {'game_version', 'base_build', 'data_version'} or None if unknown."""
    if not game_version or game_version == 'latest':
        return None
    parts = str(game_version).split('.')
    key = '.'.join(parts[:3])
    rec = known_versions().get(key)
    if rec is None:
        return None
    return {'game_version': key, **rec}


def find_sc2_binary(version=None):
    """Locate the SC2_x64 binary for a game version (or the newest install).

    `version` may be a game-version string ('4.10.0'), a base-build int, or
    None.  Version-specific SC2PATH overrides (reference
    `replay_decoder.py:375-377`): SC2PATH4_10_0 takes precedence for 4.10.0.
    """
    ver = version_info(version) if not isinstance(version, int) else None
    base_build = (ver or {}).get('base_build',
                                 version if isinstance(version, int) else None)
    sc2path = None
    if ver is not None:
        env_key = 'SC2PATH{}_{}_{}'.format(*ver['game_version'].split('.'))
        sc2path = os.environ.get(env_key)
    sc2path = sc2path or os.environ.get('SC2PATH')
    if not sc2path:
        raise FileNotFoundError('SC2PATH is not set')
    versions_dir = os.path.join(sc2path, 'Versions')
    if not os.path.isdir(versions_dir):
        raise FileNotFoundError(f'no Versions/ under {sc2path}')
    bases = sorted(d for d in os.listdir(versions_dir) if d.startswith('Base'))
    if not bases:
        raise FileNotFoundError(f'no SC2 versions under {versions_dir}')
    if base_build is not None:
        want = f'Base{base_build}'
        if want in bases:
            base = want
        else:
            # nearest installed build >= wanted (pysc2's fallback behavior)
            newer = [b for b in bases if int(b[4:]) >= int(base_build)]
            base = newer[0] if newer else bases[-1]
    else:
        base = bases[-1]
    return os.path.join(versions_dir, base, 'SC2_x64'), int(base[4:]), sc2path


def launch_game_process(cfg=None, port=None, version=None, extra_args=(),
                        verbose=False):
    """Start one SC2 process listening on a websocket port.

    Version priority: explicit `version` arg (replay-sniffed routing,
    reference `replay_decoder.py:361-380`) > cfg.env.game_version > newest
    installed.  Passes -dataVersion when the version table knows it (needed
    for replays to load assets of the right build).
    """
    port = port or pick_unused_ports(1)[0]
    cfg_version = ((cfg or {}).get('env', {}) or {}).get('game_version')
    version = version or cfg_version
    binary, base_build, sc2path = find_sc2_binary(version)
    args = [binary, '-listen', '127.0.0.1', '-port', str(port),
            '-headlessNoRender', '-displayMode', '0',
            '-dataDir', sc2path]
    ver = version_info(version)
    if ver and ver.get('data_version'):
        args += ['-dataVersion', ver['data_version'].upper()]
    args += list(extra_args)
    out = None if verbose else subprocess.DEVNULL
    proc = subprocess.Popen(args, stdout=out, stderr=out,
                            cwd=os.path.dirname(binary))
    return proc, port


class RemoteController:
    """Blocking request/response client for one SC2 process.

    Reference surface: `pysc2/lib/remote_controller.py:127-385`.  `conn`
    injection lets conformance tests drive the exact request protos through
    a fake transport.
    """

    def __init__(self, host, port, timeout=120, connect_retries=60, conn=None):
        if not SC2_PROTO_AVAILABLE:
            raise ImportError('google.protobuf unavailable; cannot speak sc2api')
        self._status = sc_pb.launched
        self._last_obs = None
        if conn is not None:
            self._ws = conn
            return
        err = None
        for _ in range(connect_retries):      # SC2 takes seconds to listen
            try:
                self._ws = WebSocket.connect(host, port, '/sc2api',
                                             timeout=timeout)
                return
            except (ConnectionError, OSError) as e:
                err = e
                time.sleep(1)
        raise ConnectionError(f'could not connect to SC2 at {host}:{port}: {err!r}')

    # ------------------------------------------------------------ plumbing
    @property
    def status(self):
        return self._status

    def _request(self, req):
        self._ws.send_binary(req.SerializeToString())
        resp = sc_pb.Response()
        resp.ParseFromString(self._ws.recv())
        if resp.HasField('status'):
            self._status = resp.status
        if resp.error:
            raise ProtocolError(f'SC2 error: {list(resp.error)}')
        return resp

    # ------------------------------------------------------- game lifecycle
    def create_game(self, map_name=None, players=None, realtime=False,
                    map_path=None, map_data=None, random_seed=None,
                    disable_fog=False):
        """players: list of {'type': 'participant'} or
        {'type': 'computer'|'bot', 'race': 'zerg', 'difficulty': 'very_hard',
        'ai_build': 'macro'} — Computer setup is what bot games need
        (reference `envs/env.py:240-246`)."""
        req = sc_pb.Request()
        cg = req.create_game
        if map_data is not None:
            cg.local_map.map_data = map_data
            if map_path:
                cg.local_map.map_path = map_path
        else:
            path = map_path or (f'{map_name}.SC2Map' if map_name else None)
            assert path, 'create_game needs map_name, map_path or map_data'
            cg.local_map.map_path = path
        for p in (players or [{'type': 'participant'}]):
            setup = cg.player_setup.add()
            ptype = p.get('type', 'participant')
            if ptype in ('computer', 'bot'):
                setup.type = sc_pb.Computer
                setup.race = RACES[p.get('race', 'random')]
                setup.difficulty = DIFFICULTIES[p.get('difficulty', 'very_hard')]
                if p.get('ai_build'):
                    setup.ai_build = AI_BUILDS[p['ai_build']]
            elif ptype == 'observer':
                setup.type = sc_pb.Observer
            else:
                setup.type = sc_pb.Participant
        cg.realtime = bool(realtime)
        cg.disable_fog = bool(disable_fog)
        if random_seed is not None:
            cg.random_seed = random_seed
        resp = self._request(req)
        if resp.create_game.HasField('error') and resp.create_game.error:
            raise ProtocolError(
                f'create_game failed: {resp.create_game.error} '
                f'{resp.create_game.error_details}')
        return resp

    @staticmethod
    def _interface(opts, raw=True, score=True, minimap_resolution=None,
                   crop_to_playable_area=True):
        """Fill an InterfaceOptions: raw + score + feature-layer minimap at
        the requested resolution (reference `envs/env.py:158-176`: width 24,
        1x1 screen resolution, minimap at map size, cropped)."""
        opts.raw = raw
        opts.score = score
        if minimap_resolution is not None:
            opts.feature_layer.width = 24
            opts.feature_layer.resolution.x = 1
            opts.feature_layer.resolution.y = 1
            opts.feature_layer.minimap_resolution.x = minimap_resolution[0]
            opts.feature_layer.minimap_resolution.y = minimap_resolution[1]
            opts.feature_layer.crop_to_playable_area = crop_to_playable_area

    def join_game(self, race, ports=None, player_name=None, raw=True,
                  score=True, minimap_resolution=None,
                  crop_to_playable_area=True):
        """`ports` (multiplayer): {'server': (game, base),
        'clients': [(game, base), ...]} — freshly reserved ports, NOT the
        websocket listen ports (reference `envs/env.py:258-266`)."""
        req = sc_pb.Request()
        jg = req.join_game
        jg.race = RACES[race] if isinstance(race, str) else race
        self._interface(jg.options, raw=raw, score=score,
                        minimap_resolution=minimap_resolution,
                        crop_to_playable_area=crop_to_playable_area)
        if player_name:
            jg.player_name = player_name
        if ports:
            jg.shared_port = 0
            jg.server_ports.game_port = ports['server'][0]
            jg.server_ports.base_port = ports['server'][1]
            for game_port, base_port in ports.get('clients', []):
                jg.client_ports.add(game_port=game_port, base_port=base_port)
        resp = self._request(req)
        if resp.join_game.HasField('error') and resp.join_game.error:
            raise ProtocolError(f'join_game failed: {resp.join_game.error} '
                                f'{resp.join_game.error_details}')
        self._player_id = resp.join_game.player_id
        return resp.join_game.player_id

    def restart_game(self):
        req = sc_pb.Request()
        req.restart_game.SetInParent()
        resp = self._request(req)
        if resp.restart_game.HasField('error') and resp.restart_game.error:
            raise ProtocolError(f'restart failed: {resp.restart_game.error}')
        return resp

    def leave(self):
        req = sc_pb.Request()
        req.leave_game.SetInParent()
        return self._request(req)

    # ------------------------------------------------------------- observe
    def observe(self, disable_fog=False, target_game_loop=0):
        req = sc_pb.Request()
        req.observation.SetInParent()
        if disable_fog:
            req.observation.disable_fog = True
        if target_game_loop:
            req.observation.game_loop = target_game_loop
        resp = self._request(req)
        o = resp.observation
        self._last_obs = o
        # action_result: plain ints (reference uses `o.result`); success == 1
        return {'raw_obs': o,
                'game_loop': o.observation.game_loop,
                'action_result': [e.result for e in o.action_errors]}

    def acts(self, action):
        """Send raw actions.  Accepts {'raw_actions': [Action...]} or a
        plain list; returns the per-action result ints."""
        raw = action.get('raw_actions', []) if isinstance(action, dict) \
            else (action or [])
        if not raw:
            return []
        req = sc_pb.Request()
        req.action.actions.extend(raw)
        resp = self._request(req)
        return list(resp.action.result)

    def act(self, action):
        return self.acts([action]) if action is not None else []

    def chat(self, message, channel=1):
        req = sc_pb.Request()
        a = req.action.actions.add()
        a.action_chat.channel = channel
        a.action_chat.message = message
        return self._request(req)

    def step(self, count=1):
        req = sc_pb.Request()
        req.step.count = count
        return self._request(req)

    # ---------------------------------------------------------- game state
    def game_info(self):
        req = sc_pb.Request()
        req.game_info.SetInParent()
        return self._request(req).game_info

    def data_raw(self, ability_id=True, unit_type_id=True, upgrade_id=True,
                 buff_id=True, effect_id=True):
        req = sc_pb.Request()
        req.data.ability_id = ability_id
        req.data.unit_type_id = unit_type_id
        req.data.upgrade_id = upgrade_id
        req.data.buff_id = buff_id
        req.data.effect_id = effect_id
        return self._request(req).data

    def data(self):
        return self.data_raw()

    def ping(self):
        req = sc_pb.Request()
        req.ping.SetInParent()
        return self._request(req).ping

    def available_maps(self):
        req = sc_pb.Request()
        req.available_maps.SetInParent()
        return self._request(req).available_maps

    def outcome(self, obs):
        """player_result -> +1/-1/0 for THIS controller's player (the list
        carries every player's result), None while the game is running."""
        results = getattr(obs['raw_obs'], 'player_result', None)
        if not results:
            return None
        pid = getattr(self, '_player_id', None)
        raw = next((r.result for r in results if r.player_id == pid),
                   results[0].result)
        mapping = {1: 1, 2: -1, 3: 0}     # Victory / Defeat / Tie
        return mapping.get(raw, 0)

    # -------------------------------------------------------------- replay
    def save_replay(self, replay_dir):
        req = sc_pb.Request()
        req.save_replay.SetInParent()
        resp = self._request(req)
        os.makedirs(replay_dir, exist_ok=True)
        path = os.path.join(replay_dir, f'replay_{int(time.time())}.SC2Replay')
        with open(path, 'wb') as f:
            f.write(resp.save_replay.data)
        return path

    def start_replay(self, replay_path, player_id, minimap_resolution=None,
                     disable_fog=False, replay_data=None):
        """`minimap_resolution`: (x, y) for the feature-layer minimap —
        pass 1 of a decode uses (1, 1), pass 2 the map size (reference
        `replay_decoder.py:229,280`)."""
        req = sc_pb.Request()
        if replay_data is not None:
            req.start_replay.replay_data = replay_data
        else:
            with open(replay_path, 'rb') as f:
                req.start_replay.replay_data = f.read()
        req.start_replay.observed_player_id = player_id
        self._interface(req.start_replay.options,
                        minimap_resolution=minimap_resolution)
        if disable_fog:
            req.start_replay.disable_fog = True
        resp = self._request(req)
        if resp.start_replay.HasField('error') and resp.start_replay.error:
            raise ProtocolError(
                f'start_replay failed: {resp.start_replay.error} '
                f'{resp.start_replay.error_details}')
        return resp

    def replay_info(self, replay_path=None, replay_data=None):
        req = sc_pb.Request()
        if replay_data is not None:
            req.replay_info.replay_data = replay_data
        else:
            with open(replay_path, 'rb') as f:
                req.replay_info.replay_data = f.read()
        return self._request(req).replay_info

    def save_map(self, map_path, map_data):
        req = sc_pb.Request()
        req.save_map.map_path = map_path
        req.save_map.map_data = map_data
        return self._request(req)

    # ---------------------------------------------------------------- quit
    def quit(self):
        req = sc_pb.Request()
        req.quit.SetInParent()
        try:
            self._ws.send_binary(req.SerializeToString())
        except (WebSocketError, OSError):
            pass
        finally:
            self._ws.close()

    def close(self):
        self.quit()
