"""SC2 protocol-layer conformance tests.

A scripted fake SC2 speaks real protobuf wire format behind the controller's
connection interface, so these tests drive `RemoteController`,
`SC2Env.reset/step` and `ReplayDecoder._parse_replay` through the FULL
request sequence — create/join/observe/act/step/game_info/start_replay —
asserting the exact request protos, with zero game binary.  The protos are
descriptor-generated (lib/sc2_protos.py), matching the reference's
`pysc2/lib/features_test.py` synthetic-ResponseObservation pattern.
"""
import json
import os
import socket
import struct
import threading

import pytest
import torch

import proto_obs as P
from distar_amd.envs import protocol
from distar_amd.envs.protocol import RemoteController, ProtocolError
from distar_amd.lib import mpq
from distar_amd.utils.ws import WebSocket, server_handshake

pb = P.pb


class FakeSC2:
    """In-memory SC2: parses each Request, records it, scripts Responses."""

    next_player_id = 1            # class-level: joins get distinct ids

    def __init__(self, end_after_observes=None, actions_on_observe=None):
        self.requests = []
        self._out = []
        self.game_loop = 0
        self.observe_count = 0
        self.end_after_observes = end_after_observes
        self.actions_on_observe = actions_on_observe or {}
        self.closed = False

    # --- conn interface used by RemoteController
    def send_binary(self, data):
        req = pb.Request()
        req.ParseFromString(data)
        self.requests.append(req)
        self._out.append(self._respond(req).SerializeToString())

    def recv(self):
        return self._out.pop(0)

    def close(self):
        self.closed = True

    # --- scripted responses
    def _respond(self, req):
        resp = pb.Response()
        if req.HasField('create_game'):
            resp.create_game.SetInParent()
            resp.status = pb.init_game
        elif req.HasField('join_game'):
            resp.join_game.player_id = FakeSC2.next_player_id
            FakeSC2.next_player_id += 1
            resp.status = pb.in_game
        elif req.HasField('observation'):
            self.observe_count += 1
            done = (self.end_after_observes is not None
                    and self.observe_count >= self.end_after_observes)
            acts = self.actions_on_observe.get(self.observe_count, [])
            ob = P.response_observation(
                game_loop=self.game_loop,
                with_result=[pb.Victory, pb.Defeat] if done else None,
                actions=acts)
            resp.observation.CopyFrom(ob)
            resp.status = pb.ended if done else pb.in_game
        elif req.HasField('action'):
            resp.action.result.extend([pb.Success] * len(req.action.actions))
        elif req.HasField('step'):
            self.game_loop += req.step.count
            resp.step.simulation_loop = self.game_loop
        elif req.HasField('game_info'):
            resp.game_info.CopyFrom(P.game_info())
        elif req.HasField('data'):
            a = resp.data.abilities.add()
            a.ability_id = 3674
            a.link_name = 'Attack'
        elif req.HasField('ping'):
            resp.ping.game_version = '4.10.0'
            resp.ping.base_build = 75689
        elif req.HasField('start_replay'):
            resp.start_replay.SetInParent()
            resp.status = pb.in_replay
            self.observe_count = 0
            self.game_loop = 0
        elif req.HasField('restart_game'):
            resp.restart_game.SetInParent()
            self.observe_count = 0
            self.game_loop = 0
        elif req.HasField('save_replay'):
            resp.save_replay.data = b'FAKE_REPLAY'
        elif req.HasField('replay_info'):
            resp.replay_info.game_version = '4.10.0.75689'
            resp.replay_info.base_build = 75689
            resp.replay_info.game_duration_loops = 1000
        elif req.HasField('save_map'):
            resp.save_map.SetInParent()
        elif req.HasField('available_maps'):
            resp.available_maps.local_map_paths.append('KingsCove.SC2Map')
        elif req.HasField('leave_game'):
            resp.leave_game.SetInParent()
            resp.status = pb.launched
        elif req.HasField('quit'):
            resp.quit.SetInParent()
            resp.status = pb.quit
        return resp


def _controller(fake):
    return RemoteController('127.0.0.1', 0, conn=fake)


# ---------------------------------------------------------------- controller
def test_controller_create_join_request_protos():
    fake = FakeSC2()
    ctrl = _controller(fake)
    ctrl.create_game(map_name='KingsCove',
                     players=[{'type': 'participant'},
                              {'type': 'computer', 'race': 'terran',
                               'difficulty': 'very_hard', 'ai_build': 'macro'}],
                     realtime=True, random_seed=42)
    req = fake.requests[-1]
    assert req.HasField('create_game')
    cg = req.create_game
    assert cg.local_map.map_path == 'KingsCove.SC2Map'
    assert cg.realtime is True and cg.random_seed == 42
    assert len(cg.player_setup) == 2
    assert cg.player_setup[0].type == pb.Participant
    assert cg.player_setup[1].type == pb.Computer
    assert cg.player_setup[1].race == pb.Terran
    assert cg.player_setup[1].difficulty == pb.VeryHard
    assert cg.player_setup[1].ai_build == 5          # macro

    player_id = ctrl.join_game(
        'zerg', ports={'server': (5001, 5002), 'clients': [(5003, 5004)]},
        minimap_resolution=(160, 152))
    assert player_id == 1
    jg = fake.requests[-1].join_game
    assert jg.race == pb.Zerg
    assert jg.options.raw and jg.options.score
    assert jg.options.feature_layer.width == 24
    assert jg.options.feature_layer.resolution.x == 1
    assert jg.options.feature_layer.minimap_resolution.x == 160
    assert jg.options.feature_layer.minimap_resolution.y == 152
    assert jg.server_ports.game_port == 5001
    assert jg.server_ports.base_port == 5002
    assert jg.client_ports[0].game_port == 5003


def test_controller_full_surface_no_attribute_errors():
    fake = FakeSC2()
    ctrl = _controller(fake)
    gi = ctrl.game_info()
    assert gi.map_name == 'KingsCove'
    assert gi.start_raw.map_size.x == 160
    data = ctrl.data()
    assert data.abilities[0].ability_id == 3674
    ping = ctrl.ping()
    assert ping.base_build == 75689
    maps = ctrl.available_maps()
    assert 'KingsCove.SC2Map' in maps.local_map_paths
    obs = ctrl.observe()
    assert obs['game_loop'] == 0 and obs['action_result'] == []
    act = P.raw_action(3674, unit_tags=[2000], target_pos=(30., 40.))
    results = ctrl.acts({'raw_actions': [act]})
    assert results == [1]                      # ints, not ActionError protos
    ctrl.step(4)
    assert fake.requests[-1].step.count == 4
    ctrl.chat('glhf')
    assert fake.requests[-1].action.actions[0].action_chat.message == 'glhf'
    ctrl.save_map('m.SC2Map', b'DATA')
    assert fake.requests[-1].save_map.map_data == b'DATA'
    ctrl.leave()
    ctrl.quit()
    assert fake.closed


def test_controller_raises_on_response_error():
    class ErrFake(FakeSC2):
        def _respond(self, req):
            resp = pb.Response()
            resp.error.append('bad request')
            return resp
    ctrl = _controller(ErrFake())
    with pytest.raises(ProtocolError):
        ctrl.step(1)


# ----------------------------------------------------------------------- env
def _patch_game_stack(monkeypatch, fakes):
    """Route launch_game_process + websocket connects to FakeSC2 instances."""
    from distar_amd.envs import env as env_mod

    FakeSC2.next_player_id = 1

    counter = {'n': 0}

    def fake_launch(cfg=None, port=None, version=None, **kw):
        class Proc:
            def kill(self):
                pass
        counter['n'] += 1
        return Proc(), 20000 + counter['n']

    def fake_connect(host, port, resource='/sc2api', timeout=120.0):
        fake = FakeSC2(end_after_observes=6)
        fakes.append(fake)
        return fake

    monkeypatch.setattr(env_mod, 'launch_game_process', fake_launch)
    monkeypatch.setattr(protocol.WebSocket, 'connect',
                        staticmethod(fake_connect))
    return counter


def test_sc2env_reset_step_full_sequence(monkeypatch):
    from distar_amd.envs.env import SC2Env
    from distar_amd.utils.config import Config
    fakes = []
    _patch_game_stack(monkeypatch, fakes)
    env = SC2Env(Config({'env': {'player_num': 2, 'map_name': 'KingsCove',
                                 'races': ['zerg', 'zerg']}}), seed=3)
    obs = env.reset()
    assert set(obs.keys()) == {0, 1}
    assert len(fakes) == 2
    host = fakes[0]
    # request sequence on the host controller: create -> join -> observe
    assert host.requests[0].HasField('create_game')
    assert host.requests[1].HasField('join_game')
    assert host.requests[-1].HasField('observation')
    # the client controller joined but did not create
    assert fakes[1].requests[0].HasField('join_game')
    # game/base ports are freshly reserved — NOT the (fake) websocket ports
    jg = host.requests[1].join_game
    ws_ports = {20001, 20002}
    game_ports = {jg.server_ports.game_port, jg.server_ports.base_port,
                  jg.client_ports[0].game_port, jg.client_ports[0].base_port}
    assert len(game_ports) == 4
    assert not (game_ports & ws_ports)
    # both agents got the same port set
    assert fakes[1].requests[0].join_game.server_ports.game_port == \
        jg.server_ports.game_port

    act = P.raw_action(3674, unit_tags=[2000], target_pos=(30., 40.))
    obs, rewards, done, infos = env.step(
        {0: {'raw_actions': [act], 'skip_steps': 2},
         1: {'raw_actions': [], 'skip_steps': 0}})
    # action forwarded verbatim, then stepped >= 1 loops on every controller
    action_reqs = [r for r in host.requests if r.HasField('action')]
    assert action_reqs[-1].action.actions[0].action_raw.unit_command.ability_id \
        == 3674
    step_reqs = [r for r in host.requests if r.HasField('step')]
    assert step_reqs and step_reqs[-1].step.count >= 1
    assert fakes[1].requests[-1].HasField('observation')
    env.close()
    assert all(f.closed for f in fakes)


def test_sc2env_bot_game_setup(monkeypatch):
    from distar_amd.envs.env import SC2Env
    from distar_amd.utils.config import Config
    fakes = []
    _patch_game_stack(monkeypatch, fakes)
    env = SC2Env(Config({'env': {'player_num': 2, 'map_name': 'KingsCove',
                                 'realtime': True,
                                 'bot_difficulty': 'very_hard',
                                 'bot_race': 'terran'}}), seed=1)
    env.reset()
    assert len(fakes) == 1                     # one agent, one SC2 process
    cg = fakes[0].requests[0].create_game
    assert cg.realtime is True
    assert len(cg.player_setup) == 2
    assert cg.player_setup[0].type == pb.Participant
    assert cg.player_setup[1].type == pb.Computer
    assert cg.player_setup[1].difficulty == pb.VeryHard
    # single-agent game: no port set in join
    jg = fakes[0].requests[1].join_game
    assert not jg.HasField('server_ports')
    # realtime: no step requests issued
    env.step({0: {'raw_actions': [], 'skip_steps': 0}})
    assert not any(r.HasField('step') for r in fakes[0].requests)
    env.close()


# ------------------------------------------------------------ replay decoder
def _synthetic_replay(tmp_path, game_version='4.10.0.75689',
                      base_build=75689):
    """Build a real (minimal, v1) MPQ archive holding
    replay.gamemetadata.json, exercising lib/mpq end to end."""
    meta = json.dumps({'GameVersion': game_version,
                       'BaseBuild': f'Base{base_build}',
                       'DataBuild': f'Base{base_build}'}).encode()
    fname = 'replay.gamemetadata.json'
    header_offset = 512
    file_offset = 32                             # relative to header
    ht_entries, bt_entries = 4, 1
    ht_off = file_offset + len(meta)
    ht_off += (-ht_off) % 16
    bt_off = ht_off + ht_entries * 16
    # hash table: 1 real entry at the probe slot, rest empty
    empty = (0xFFFFFFFF, 0xFFFFFFFF, 0xFFFFFFFF, 0xFFFFFFFF)
    slot = mpq.mpq_hash(fname, mpq.HASH_TABLE_OFFSET) & (ht_entries - 1)
    entries = [empty] * ht_entries
    entries[slot] = (mpq.mpq_hash(fname, mpq.HASH_NAME_A),
                     mpq.mpq_hash(fname, mpq.HASH_NAME_B), 0, 0)
    ht_raw = b''.join(struct.pack('<4I', *e) for e in entries)
    bt_raw = struct.pack('<4I', file_offset, len(meta), len(meta),
                         mpq.FLAG_EXISTS)
    ht_enc = mpq.encrypt(ht_raw, mpq.mpq_hash('(hash table)',
                                              mpq.HASH_FILE_KEY))
    bt_enc = mpq.encrypt(bt_raw, mpq.mpq_hash('(block table)',
                                              mpq.HASH_FILE_KEY))
    archive_size = bt_off + len(bt_raw)
    header = mpq.MPQ_HEADER_MAGIC + struct.pack(
        '<IIHHIIII', 32, archive_size, 0, 3, ht_off, bt_off,
        ht_entries, bt_entries)
    user = mpq.MPQ_USER_DATA_MAGIC + struct.pack('<III', 512, header_offset, 0)
    blob = bytearray(header_offset + archive_size)
    blob[0:len(user)] = user
    blob[header_offset:header_offset + 32] = header
    blob[header_offset + file_offset:header_offset + file_offset + len(meta)] \
        = meta
    blob[header_offset + ht_off:header_offset + ht_off + len(ht_enc)] = ht_enc
    blob[header_offset + bt_off:header_offset + bt_off + len(bt_enc)] = bt_enc
    path = os.path.join(tmp_path, 'test.SC2Replay')
    with open(path, 'wb') as f:
        f.write(bytes(blob))
    return path


def test_mpq_reader_roundtrip(tmp_path):
    path = _synthetic_replay(str(tmp_path))
    meta = json.loads(mpq.MPQArchive(path).read_file(
        'replay.gamemetadata.json'))
    assert meta['GameVersion'] == '4.10.0.75689'
    assert meta['BaseBuild'] == 'Base75689'


def test_replay_decoder_two_pass_conformance(tmp_path, monkeypatch):
    from distar_amd.data.replay_decoder import ReplayDecoder
    from distar_amd.utils.config import Config
    replay = _synthetic_replay(str(tmp_path))

    launched = {}
    fakes = []

    def fake_launch(cfg=None, port=None, version=None, **kw):
        launched['version'] = version
        class Proc:
            def kill(self):
                pass
        return Proc(), 23456

    act1 = P.raw_action(3674, unit_tags=[2000], target_pos=(30., 40.),
                        game_loop=10)

    def fake_connect(host, port, resource='/sc2api', timeout=120.0):
        fake = FakeSC2(end_after_observes=2, actions_on_observe={1: [act1]})
        fakes.append(fake)
        return fake

    from distar_amd.data import replay_decoder as rd_mod
    monkeypatch.setattr(rd_mod, 'launch_game_process', fake_launch)
    monkeypatch.setattr(protocol.WebSocket, 'connect',
                        staticmethod(fake_connect))

    dec = ReplayDecoder(Config({'env': {}}))
    traj = dec.run(replay, player_idx=0)
    # the sniffed version routed the binary selection
    assert launched['version'] == '4.10.0'
    fake = fakes[0]
    kinds = []
    for r in fake.requests:
        for f in ('start_replay', 'observation', 'step', 'game_info',
                  'action'):
            if r.HasField(f):
                kinds.append(f)
    # pass 1: start_replay @1x1 -> game_info -> observe/step loop
    assert kinds[0] == 'start_replay'
    sr1 = fake.requests[0].start_replay
    assert sr1.observed_player_id == 1
    assert sr1.options.feature_layer.minimap_resolution.x == 1
    assert kinds[1] == 'game_info'
    # pass 2 re-opens at map resolution
    second_sr = [r.start_replay for r in fake.requests[2:]
                 if r.HasField('start_replay')]
    assert second_sr, 'pass 2 start_replay missing'
    assert second_sr[0].options.feature_layer.minimap_resolution.x == 160
    assert second_sr[0].options.feature_layer.minimap_resolution.y == 152
    # decode produced a transformed step with the action labels attached
    assert traj is not None and len(traj) == 1
    step = traj[0]
    assert step['action_info']['action_type'] is not None
    assert 'spatial_info' in step and 'entity_info' in step
    assert torch.is_tensor(step['selected_units_num'])
    dec.close()
    assert fake.closed


# ----------------------------------------------------- versioned binary pick
def test_launch_selects_versioned_binary(tmp_path, monkeypatch):
    sc2 = tmp_path / 'SC2'
    for build in (70154, 75689, 81433):
        d = sc2 / 'Versions' / f'Base{build}'
        d.mkdir(parents=True)
        (d / 'SC2_x64').write_text('')
    monkeypatch.setenv('SC2PATH', str(sc2))

    calls = {}

    class FakeProc:
        def __init__(self, args, **kw):
            calls['args'] = args

    monkeypatch.setattr(protocol.subprocess, 'Popen', FakeProc)
    proc, port = protocol.launch_game_process(version='4.10.0', port=12345)
    args = calls['args']
    assert f'Base75689{os.sep}SC2_x64' in args[0]
    assert '-dataVersion' in args
    assert args[args.index('-dataVersion') + 1] == \
        'B89B5D6FA7CBF6452E721311BFBC6CB2'
    # base-build routing (unknown point release sniffed to a build int)
    protocol.launch_game_process(version=81433, port=12345)
    assert f'Base81433{os.sep}SC2_x64' in calls['args'][0]
    # no version: newest install
    protocol.launch_game_process(port=12345)
    assert f'Base81433{os.sep}SC2_x64' in calls['args'][0]


# ----------------------------------------------------------- websocket layer
def test_ws_client_loopback_binary_roundtrip():
    srv = socket.socket()
    srv.bind(('127.0.0.1', 0))
    srv.listen(1)
    port = srv.getsockname()[1]

    def serve():
        conn, _ = srv.accept()
        ws = server_handshake(conn)
        msg = ws.recv()
        ws.send_binary(b'echo:' + msg)
        ws.close()

    t = threading.Thread(target=serve)
    t.start()
    c = WebSocket.connect('127.0.0.1', port)
    payload = os.urandom(200000)            # 64-bit length frame path
    c.send_binary(payload)
    assert c.recv() == b'echo:' + payload
    c.close()
    t.join(timeout=10)
    srv.close()


# -------------------------------------------- features on descriptor protos
def test_transform_obs_on_real_protos():
    """`Features.transform_obs` + `reverse_raw_action` against REAL
    descriptor-generated ResponseObservation/ResponseGameInfo messages
    (not duck-typed namespaces) — field-name drift fails here."""
    from distar_amd.lib.features import Features
    feat = Features(P.game_info(), P.response_observation())
    obs = P.response_observation(num_units=10, upgrades=(76,))
    step = feat.transform_obs(obs, padding_spatial=True)
    assert step['spatial_info']['height_map'].shape[-2:] == (152, 160)
    assert int(step['entity_num']) == 10
    assert step['entity_info']['unit_type'].shape[0] == 10
    tags = step['game_info']['tags']
    assert 1000 in tags and 2000 in tags
    act = P.raw_action(3674, unit_tags=[2000], target_pos=(30., 40.))
    action_info, mask, su_num, *_rest, invalid = feat.reverse_raw_action(
        act, tags)
    assert not invalid
    assert action_info['action_type'] is not None
    assert int(su_num) == 2      # one selected unit + the end flag


def test_gen_z_live_decode_end_to_end(tmp_path, monkeypatch):
    """The gen_z live path over the fake-websocket SC2: decode, winner
    filter, Z extraction, and map/race/born aggregation (reference
    bin/gen_z.py worker_loop+result_loop)."""
    from distar_amd.bin.gen_z import aggregate, replay_entries
    replay = _synthetic_replay(str(tmp_path))

    act1 = P.raw_action(3674, unit_tags=[2000], target_pos=(30., 40.),
                        game_loop=10)

    def fake_launch(cfg=None, port=None, version=None, **kw):
        class Proc:
            def kill(self):
                pass
        return Proc(), 23456

    def fake_connect(host, port, resource='/sc2api', timeout=120.0):
        # player 1 wins, player 2 loses -> only one side kept
        return FakeSC2(end_after_observes=2, actions_on_observe={1: [act1]})

    from distar_amd.data import replay_decoder as rd_mod
    monkeypatch.setattr(rd_mod, 'launch_game_process', fake_launch)
    monkeypatch.setattr(protocol.WebSocket, 'connect',
                        staticmethod(fake_connect))

    entries = replay_entries(str(tmp_path), min_loop=0)
    # both sides decoded; FakeSC2 reports Victory for player 1 only
    assert len(entries) == 1
    map_name, mix, born, z = entries[0]
    assert map_name == 'KingsCove'
    assert mix == 'zerg'
    bo, cum_idx, bo_loc, end_loop = z
    assert isinstance(bo, list) and isinstance(cum_idx, list)
    agg = aggregate(entries)
    assert 'KingsCove' in agg and 'zerg' in agg['KingsCove']


@pytest.mark.timeout(600)
def test_play_path_agent_over_fake_sc2(monkeypatch):
    """The play/eval path end to end over the fake-websocket SC2: real
    SC2Env (create/join/observe/act/step protos) + the real Agent + model
    inference, including transform_action routing of the agent's func_id
    action dicts into raw-action protos (reference `envs/env.py:457-483`,
    `bin/play.py`)."""
    from distar_amd.actor.agent import Agent
    from distar_amd.envs.env import SC2Env
    from distar_amd.utils.config import Config
    fakes = []
    _patch_game_stack(monkeypatch, fakes)
    cfg = Config({'common': {'type': 'play'},
                  'actor': {'traj_len': 3, 'job_type': 'eval_test'},
                  'env': {'player_num': 2, 'map_name': 'KingsCove',
                          'races': ['zerg', 'zerg']},
                  'agent': {}})
    env = SC2Env(cfg, seed=7)
    obs = env.reset()
    assert 'game_info_proto' in obs[0]
    agents = [Agent(cfg, env_id=0) for _ in range(2)]
    for i, agent in enumerate(agents):
        agent.player_id = f'MP{i}'
        agent.reset(obs=obs.get(i))
    import torch as _t
    with _t.no_grad():
        for _ in range(2):
            actions = {i: agents[i].step(obs[i])[0] for i in obs}
            for a in actions.values():
                assert 'func_id' in a
            obs2, rewards, done, infos = env.step(actions)
            obs.update(obs2)
            if done:
                break
    # some controller received a RequestAction with a raw action whenever
    # the agent emitted a non-no-op
    acted = any(r.HasField('action') and len(r.action.actions) > 0
                for f in fakes for r in f.requests)
    stepped = any(r.HasField('step') for f in fakes for r in f.requests)
    assert stepped
    env.close()


def test_replay_actor_worker_decodes_and_pushes(tmp_path, monkeypatch):
    """ReplayActor worker over the fake SC2: decodes each (replay, player)
    and Adapter.pushes the step lists (reference replay_actor.py:10-73)."""
    from distar_amd.data import replay_actor as ra_mod
    from distar_amd.data import replay_decoder as rd_mod
    from distar_amd.utils.config import Config
    replay = _synthetic_replay(str(tmp_path))

    act1 = P.raw_action(3674, unit_tags=[2000], target_pos=(30., 40.),
                        game_loop=10)

    def fake_launch(cfg=None, port=None, version=None, **kw):
        class Proc:
            def kill(self):
                pass
        return Proc(), 23456

    def fake_connect(host, port, resource='/sc2api', timeout=120.0):
        return FakeSC2(end_after_observes=2, actions_on_observe={1: [act1]})

    monkeypatch.setattr(rd_mod, 'launch_game_process', fake_launch)
    monkeypatch.setattr(protocol.WebSocket, 'connect',
                        staticmethod(fake_connect))

    pushes = []

    class FakeAdapter:
        def __init__(self, cfg):
            pass

        def push(self, data, token=None, fs_type=None):
            pushes.append((token, fs_type, len(data)))

    monkeypatch.setattr(ra_mod, 'Adapter', FakeAdapter)
    ra_mod._worker(Config({'env': {}}), [replay], 0)
    # both players decoded successfully (1 transformed step each)
    assert len(pushes) == 2
    assert all(t == 'replay' and fs == 'nppickle' and n == 1
               for t, fs, n in pushes)


@pytest.mark.timeout(900)
def test_actor_rollout_over_fake_sc2(monkeypatch):
    """The full Actor episode loop with env_type='sc2' (the bin/play.py
    composition) over the fake-websocket SC2: job setup, agent inference,
    transform_action protos, episode termination and result extraction."""
    import torch as _t
    from distar_amd.actor.actor import Actor
    from distar_amd.utils.config import Config
    fakes = []
    _patch_game_stack(monkeypatch, fakes)
    _t.manual_seed(0)
    cfg = Config({'actor': {'episode_num': 1, 'traj_len': 4,
                            'env_type': 'sc2'},
                  'env': {'player_num': 2, 'map_name': 'KingsCove',
                          'races': ['zerg', 'zerg']},
                  'common': {'experiment_name': 'test_actor_sc2',
                             'type': 'train'}})
    actor = Actor(cfg)
    with _t.no_grad():
        results = actor.run()
    assert len(results) == 1
    r = results[0]
    # FakeSC2 scripts Victory for player 1, Defeat for player 2
    assert r['0']['winloss'] == 1 and r['1']['winloss'] == -1
    # the env spoke real protos: create+join+step traffic happened
    assert any(req.HasField('create_game')
               for f in fakes for req in f.requests)
    assert any(req.HasField('step') for f in fakes for req in f.requests)
