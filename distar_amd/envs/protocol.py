"""StarCraft II protocol client (reference: the vendored pysc2's
`run_configs` + `lib/remote_controller.py:127-350`).

Everything protobuf-typed is gated on `s2clientprotocol` availability (this
offline image ships neither the bindings nor the game binary).  The module
still defines the full controller surface so the env/actor/replay layers
compile and their control flow is testable against `MockController`.
"""
import os
import subprocess

try:
    from s2clientprotocol import sc2api_pb2 as sc_pb          # noqa: F401
    from s2clientprotocol import common_pb2 as sc_common      # noqa: F401
    import websocket                                          # noqa: F401
    SC2_PROTO_AVAILABLE = True
except ImportError:
    SC2_PROTO_AVAILABLE = False

RACES = {'zerg': 2, 'terran': 1, 'protoss': 3, 'random': 4}


def find_sc2_binary(version=None):
    sc2path = os.environ.get('SC2PATH')
    if not sc2path:
        raise FileNotFoundError('SC2PATH is not set')
    versions_dir = os.path.join(sc2path, 'Versions')
    if version:
        base = os.path.join(versions_dir, f'Base{version}')
    else:
        bases = sorted(d for d in os.listdir(versions_dir) if d.startswith('Base'))
        if not bases:
            raise FileNotFoundError(f'no SC2 versions under {versions_dir}')
        base = os.path.join(versions_dir, bases[-1])
    return os.path.join(base, 'SC2_x64')


def launch_game_process(cfg, port=None):
    """Start one SC2 process listening on a websocket port."""
    from ..utils.http import pick_unused_port
    port = port or pick_unused_port()
    binary = find_sc2_binary((cfg or {}).get('env', {}).get('game_version'))
    proc = subprocess.Popen(
        [binary, '-listen', '127.0.0.1', '-port', str(port), '-headlessNoRender'],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    return proc, port


class RemoteController:
    """Blocking websocket request/response client for one SC2 process."""

    def __init__(self, host, port, timeout=120):
        if not SC2_PROTO_AVAILABLE:
            raise ImportError('s2clientprotocol not available')
        import websocket as ws
        self._ws = ws.create_connection(
            f'ws://{host}:{port}/sc2api', timeout=timeout)

    def _request(self, req):
        from s2clientprotocol import sc2api_pb2 as sc_pb
        self._ws.send_binary(req.SerializeToString())
        resp = sc_pb.Response()
        resp.ParseFromString(self._ws.recv())
        if resp.error:
            raise ConnectionError(f'SC2 error: {resp.error}')
        return resp

    def create_game(self, map_name, agent_num, ports):
        from s2clientprotocol import sc2api_pb2 as sc_pb
        req = sc_pb.Request()
        req.create_game.local_map.map_path = f'{map_name}.SC2Map'
        for _ in range(agent_num):
            req.create_game.player_setup.add(type=sc_pb.Participant)
        req.create_game.realtime = False
        return self._request(req)

    def join_game(self, race, ports):
        from s2clientprotocol import sc2api_pb2 as sc_pb
        req = sc_pb.Request()
        req.join_game.race = RACES[race]
        req.join_game.options.raw = True
        req.join_game.options.score = True
        if len(ports) > 1:
            req.join_game.server_ports.game_port = ports[0]
            req.join_game.server_ports.base_port = ports[0] + 1
            for p in ports[1:]:
                cp = req.join_game.client_ports.add()
                cp.game_port = p
                cp.base_port = p + 1
        return self._request(req)

    def observe(self):
        from s2clientprotocol import sc2api_pb2 as sc_pb
        req = sc_pb.Request()
        req.observation.SetInParent()
        resp = self._request(req)
        return {'raw_obs': resp.observation,
                'game_loop': resp.observation.observation.game_loop,
                'action_result': [r for r in
                                  getattr(resp.observation, 'action_errors', [])]}

    def acts(self, action):
        from s2clientprotocol import sc2api_pb2 as sc_pb
        req = sc_pb.Request()
        req.action.actions.extend(action['raw_actions'])
        return self._request(req)

    def step(self, count=1):
        from s2clientprotocol import sc2api_pb2 as sc_pb
        req = sc_pb.Request()
        req.step.count = count
        return self._request(req)

    @staticmethod
    def outcome(obs):
        """player_result -> +1/-1/0, None while the game is running."""
        results = getattr(obs['raw_obs'], 'player_result', None)
        if not results:
            return None
        mapping = {1: 1, 2: -1, 3: 0}     # Victory / Defeat / Tie
        return mapping.get(results[0].result, 0)

    def restart_game(self):
        from s2clientprotocol import sc2api_pb2 as sc_pb
        req = sc_pb.Request()
        req.restart_game.SetInParent()
        return self._request(req)

    def save_replay(self, replay_dir):
        from s2clientprotocol import sc2api_pb2 as sc_pb
        import time as _time
        req = sc_pb.Request()
        req.save_replay.SetInParent()
        resp = self._request(req)
        path = os.path.join(replay_dir, f'replay_{int(_time.time())}.SC2Replay')
        with open(path, 'wb') as f:
            f.write(resp.save_replay.data)
        return path

    def start_replay(self, replay_path, player_id, resolution=1):
        from s2clientprotocol import sc2api_pb2 as sc_pb
        req = sc_pb.Request()
        with open(replay_path, 'rb') as f:
            req.start_replay.replay_data = f.read()
        req.start_replay.observed_player_id = player_id
        req.start_replay.options.raw = True
        return self._request(req)

    def quit(self):
        from s2clientprotocol import sc2api_pb2 as sc_pb
        req = sc_pb.Request()
        req.quit.SetInParent()
        try:
            self._request(req)
        finally:
            self._ws.close()
