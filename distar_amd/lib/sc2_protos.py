"""Minimal wire-compatible s2clientprotocol message definitions.

The offline image ships neither the `s2clientprotocol` package nor the SC2
binary, but the SC2-facing layers (envs/protocol.py, envs/env.py,
data/replay_decoder.py) must be real, runnable code — not import-gated
scaffolding.  This module builds the subset of Blizzard's public
s2client-proto messages that those layers actually read/write, as REAL
google.protobuf messages (dynamic descriptors), so:

  * request/response serialization is genuine protobuf wire format,
  * conformance tests can drive the full request sequence through a fake
    websocket and parse the exact bytes the controller sent,
  * on a machine with the official `s2clientprotocol` installed, that
    package is preferred automatically (`get_protos()`), making this module
    a drop-in fallback.

Field numbers are transcribed from Blizzard's published s2client-proto
(sc2api.proto / raw.proto / common.proto / score.proto / data.proto).  Only
messages and fields consumed by this repo are declared; proto2 `optional`
semantics are used so `HasField` works as the consumers expect.  `oneof`
groups are declared as plain optionals (serialisation is identical; the
auto-clear-sibling behaviour is not needed by this stack).

Reference surface being replaced: the vendored pysc2's use of the official
package (`distar/pysc2/lib/remote_controller.py:127-350`).
"""
import threading

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_F = descriptor_pb2.FieldDescriptorProto
_TYPES = {
    'i32': _F.TYPE_INT32,
    'u32': _F.TYPE_UINT32,
    'i64': _F.TYPE_INT64,
    'u64': _F.TYPE_UINT64,
    'f32': _F.TYPE_FLOAT,
    'bool': _F.TYPE_BOOL,
    'str': _F.TYPE_STRING,
    'bytes': _F.TYPE_BYTES,
}

_PKG = 'DistarSC2'

# (message name, [(field name, number, type[, 'rep'])]) — type 'M:Name' is a
# message-typed field.  Numbers are the wire contract; see module docstring.
_MESSAGES = [
    # ----------------------------------------------------- common.proto
    ('Point2D', [('x', 1, 'f32'), ('y', 2, 'f32')]),
    ('Point', [('x', 1, 'f32'), ('y', 2, 'f32'), ('z', 3, 'f32')]),
    ('PointI', [('x', 1, 'i32'), ('y', 2, 'i32')]),
    ('Size2DI', [('x', 1, 'i32'), ('y', 2, 'i32')]),
    ('RectangleI', [('p0', 1, 'M:PointI'), ('p1', 2, 'M:PointI')]),
    ('ImageData', [('bits_per_pixel', 1, 'i32'), ('size', 2, 'M:Size2DI'),
                   ('data', 3, 'bytes')]),
    # -------------------------------------------------------- raw.proto
    ('StartRaw', [('map_size', 1, 'M:Size2DI'),
                  ('pathing_grid', 2, 'M:ImageData'),
                  ('terrain_height', 3, 'M:ImageData'),
                  ('placement_grid', 4, 'M:ImageData'),
                  ('playable_area', 5, 'M:RectangleI'),
                  ('start_locations', 6, 'M:Point2D', 'rep')]),
    ('PowerSource', [('pos', 1, 'M:Point'), ('radius', 2, 'f32'),
                     ('tag', 3, 'u64')]),
    ('PlayerRaw', [('power_sources', 1, 'M:PowerSource', 'rep'),
                   ('camera', 2, 'M:Point'),
                   ('upgrade_ids', 3, 'u32', 'rep')]),
    ('UnitOrder', [('ability_id', 1, 'u32'),
                   ('target_world_space_pos', 2, 'M:Point'),
                   ('target_unit_tag', 3, 'u64'),
                   ('progress', 4, 'f32')]),
    ('PassengerUnit', [('tag', 1, 'u64'), ('health', 2, 'f32'),
                       ('health_max', 3, 'f32'), ('shield', 4, 'f32'),
                       ('energy', 5, 'f32'), ('unit_type', 6, 'u32'),
                       ('shield_max', 7, 'f32'), ('energy_max', 8, 'f32')]),
    ('RallyTarget', [('point', 1, 'M:Point'), ('tag', 2, 'u64')]),
    ('Unit', [('display_type', 1, 'i32'), ('alliance', 2, 'i32'),
              ('tag', 3, 'u64'), ('unit_type', 4, 'u32'), ('owner', 5, 'i32'),
              ('pos', 6, 'M:Point'), ('facing', 7, 'f32'),
              ('radius', 8, 'f32'), ('build_progress', 9, 'f32'),
              ('cloak', 10, 'i32'), ('is_selected', 11, 'bool'),
              ('is_on_screen', 12, 'bool'), ('is_blip', 13, 'bool'),
              ('health', 14, 'f32'), ('health_max', 15, 'f32'),
              ('shield', 16, 'f32'), ('energy', 17, 'f32'),
              ('mineral_contents', 18, 'i32'), ('vespene_contents', 19, 'i32'),
              ('is_flying', 20, 'bool'), ('is_burrowed', 21, 'bool'),
              ('orders', 22, 'M:UnitOrder', 'rep'), ('add_on_tag', 23, 'u64'),
              ('passengers', 24, 'M:PassengerUnit', 'rep'),
              ('cargo_space_taken', 25, 'i32'), ('cargo_space_max', 26, 'i32'),
              ('buff_ids', 27, 'u32', 'rep'),
              ('assigned_harvesters', 28, 'i32'),
              ('ideal_harvesters', 29, 'i32'), ('weapon_cooldown', 30, 'f32'),
              ('detect_range', 31, 'f32'), ('radar_range', 32, 'f32'),
              ('engaged_target_tag', 34, 'u64'), ('is_powered', 35, 'bool'),
              ('shield_max', 36, 'f32'), ('energy_max', 37, 'f32'),
              ('is_hallucination', 38, 'bool'), ('is_active', 39, 'bool'),
              ('attack_upgrade_level', 40, 'i32'),
              ('armor_upgrade_level', 41, 'i32'),
              ('shield_upgrade_level', 42, 'i32'),
              ('buff_duration_remain', 43, 'i32'),
              ('buff_duration_max', 44, 'i32'),
              ('rally_targets', 45, 'M:RallyTarget', 'rep')]),
    ('MapState', [('visibility', 1, 'M:ImageData'),
                  ('creep', 2, 'M:ImageData')]),
    ('Event', [('dead_units', 1, 'u64', 'rep')]),
    ('Effect', [('effect_id', 1, 'u32'), ('pos', 2, 'M:Point2D', 'rep'),
                ('alliance', 3, 'i32'), ('owner', 4, 'i32'),
                ('radius', 5, 'f32')]),
    ('RadarRing', [('pos', 1, 'M:Point'), ('radius', 2, 'f32')]),
    ('ObservationRaw', [('player', 1, 'M:PlayerRaw'),
                        ('units', 2, 'M:Unit', 'rep'),
                        ('map_state', 3, 'M:MapState'),
                        ('event', 4, 'M:Event'),
                        ('effects', 5, 'M:Effect', 'rep'),
                        ('radar', 6, 'M:RadarRing', 'rep')]),
    ('ActionRawUnitCommand', [('ability_id', 1, 'i32'),
                              ('target_world_space_pos', 2, 'M:Point2D'),
                              ('target_unit_tag', 3, 'u64'),
                              ('unit_tags', 4, 'u64', 'rep'),
                              ('queue_command', 5, 'bool')]),
    ('ActionRawCameraMove', [('center_world_space', 1, 'M:Point')]),
    ('ActionRawToggleAutocast', [('ability_id', 1, 'i32'),
                                 ('unit_tags', 2, 'u64', 'rep')]),
    ('ActionRaw', [('unit_command', 1, 'M:ActionRawUnitCommand'),
                   ('camera_move', 2, 'M:ActionRawCameraMove'),
                   ('toggle_autocast', 3, 'M:ActionRawToggleAutocast')]),
    # ------------------------------------------------------ score.proto
    ('CategoryScoreDetails', [('none', 1, 'f32'), ('army', 2, 'f32'),
                              ('economy', 3, 'f32'), ('technology', 4, 'f32'),
                              ('upgrade', 5, 'f32')]),
    ('ScoreDetails', [('idle_production_time', 1, 'f32'),
                      ('idle_worker_time', 2, 'f32'),
                      ('total_value_units', 3, 'f32'),
                      ('total_value_structures', 4, 'f32'),
                      ('killed_value_units', 5, 'f32'),
                      ('killed_value_structures', 6, 'f32'),
                      ('collected_minerals', 7, 'f32'),
                      ('collected_vespene', 8, 'f32'),
                      ('collection_rate_minerals', 9, 'f32'),
                      ('collection_rate_vespene', 10, 'f32'),
                      ('spent_minerals', 11, 'f32'),
                      ('spent_vespene', 12, 'f32'),
                      ('food_used', 13, 'M:CategoryScoreDetails'),
                      ('killed_minerals', 14, 'M:CategoryScoreDetails'),
                      ('killed_vespene', 15, 'M:CategoryScoreDetails'),
                      ('lost_minerals', 16, 'M:CategoryScoreDetails'),
                      ('lost_vespene', 17, 'M:CategoryScoreDetails')]),
    ('Score', [('score_type', 6, 'i32'), ('score', 7, 'i32'),
               ('score_details', 8, 'M:ScoreDetails')]),
    # ------------------------------------------------------- data.proto
    ('AbilityData', [('ability_id', 1, 'u32'), ('link_name', 2, 'str'),
                     ('link_index', 3, 'u32'), ('button_name', 4, 'str'),
                     ('friendly_name', 5, 'str'), ('hotkey', 6, 'str'),
                     ('remaps_to_ability_id', 7, 'u32')]),
    ('UnitTypeData', [('unit_id', 1, 'u32'), ('name', 2, 'str'),
                      ('available', 3, 'bool'), ('cargo_size', 4, 'u32')]),
    ('UpgradeData', [('upgrade_id', 1, 'u32'), ('name', 2, 'str')]),
    ('BuffData', [('buff_id', 1, 'u32'), ('name', 2, 'str')]),
    ('EffectData', [('effect_id', 1, 'u32'), ('name', 2, 'str'),
                    ('friendly_name', 3, 'str'), ('radius', 4, 'f32')]),
    # ---------------------------------------------------- spatial.proto
    ('FeatureLayersMinimap', [('height_map', 1, 'M:ImageData'),
                              ('visibility_map', 2, 'M:ImageData'),
                              ('creep', 3, 'M:ImageData'),
                              ('camera', 4, 'M:ImageData'),
                              ('player_id', 5, 'M:ImageData'),
                              ('player_relative', 6, 'M:ImageData'),
                              ('selected', 7, 'M:ImageData'),
                              ('unit_type', 8, 'M:ImageData'),
                              ('alerts', 9, 'M:ImageData'),
                              ('buildable', 10, 'M:ImageData'),
                              ('pathable', 11, 'M:ImageData')]),
    ('ObservationFeatureLayer', [('minimap_renders', 2,
                                  'M:FeatureLayersMinimap')]),
    # ----------------------------------------------------- sc2api.proto
    ('SpatialCameraSetup', [('width', 1, 'f32'),
                            ('resolution', 2, 'M:Size2DI'),
                            ('minimap_resolution', 3, 'M:Size2DI'),
                            ('crop_to_playable_area', 4, 'bool'),
                            ('allow_cheating_layers', 5, 'bool')]),
    ('LocalMap', [('map_path', 1, 'str'), ('map_data', 7, 'bytes')]),
    ('PlayerSetup', [('type', 1, 'i32'), ('race', 2, 'i32'),
                     ('difficulty', 3, 'i32'), ('player_name', 4, 'str'),
                     ('ai_build', 5, 'i32')]),
    ('RequestCreateGame', [('local_map', 1, 'M:LocalMap'),
                           ('battlenet_map_name', 2, 'str'),
                           ('player_setup', 3, 'M:PlayerSetup', 'rep'),
                           ('disable_fog', 4, 'bool'),
                           ('random_seed', 5, 'u32'),
                           ('realtime', 6, 'bool')]),
    ('ResponseCreateGame', [('error', 1, 'i32'), ('error_details', 2, 'str')]),
    ('PortSet', [('game_port', 1, 'i32'), ('base_port', 2, 'i32')]),
    ('InterfaceOptions', [('raw', 1, 'bool'), ('score', 2, 'bool'),
                          ('feature_layer', 3, 'M:SpatialCameraSetup'),
                          ('show_cloaked', 5, 'bool'),
                          ('raw_affects_selection', 6, 'bool'),
                          ('raw_crop_to_playable_area', 7, 'bool'),
                          ('show_placeholders', 8, 'bool'),
                          ('show_burrowed_shadows', 9, 'bool')]),
    ('RequestJoinGame', [('race', 1, 'i32'), ('observed_player_id', 2, 'u32'),
                         ('options', 3, 'M:InterfaceOptions'),
                         ('server_ports', 4, 'M:PortSet'),
                         ('client_ports', 5, 'M:PortSet', 'rep'),
                         ('shared_port', 6, 'i32'),
                         ('player_name', 7, 'str'), ('host_ip', 8, 'str')]),
    ('ResponseJoinGame', [('player_id', 1, 'u32'), ('error', 2, 'i32'),
                          ('error_details', 3, 'str')]),
    ('RequestRestartGame', []),
    ('ResponseRestartGame', [('error', 1, 'i32'), ('error_details', 2, 'str'),
                             ('need_hard_reset', 3, 'bool')]),
    ('RequestStartReplay', [('replay_data', 1, 'bytes'),
                            ('replay_path', 2, 'str'),
                            ('observed_player_id', 3, 'i32'),
                            ('options', 4, 'M:InterfaceOptions'),
                            ('disable_fog', 5, 'bool'),
                            ('map_data', 6, 'bytes'),
                            ('realtime', 7, 'bool'),
                            ('record_replay', 8, 'bool')]),
    ('ResponseStartReplay', [('error', 1, 'i32'), ('error_details', 2, 'str')]),
    ('RequestLeaveGame', []),
    ('ResponseLeaveGame', []),
    ('RequestQuit', []),
    ('ResponseQuit', []),
    ('RequestGameInfo', []),
    ('PlayerInfo', [('player_id', 1, 'u32'), ('type', 2, 'i32'),
                    ('race_requested', 3, 'i32'), ('race_actual', 4, 'i32'),
                    ('difficulty', 5, 'i32'), ('player_name', 6, 'str'),
                    ('ai_build', 7, 'i32')]),
    ('ResponseGameInfo', [('map_name', 1, 'str'),
                          ('local_map_path', 2, 'str'),
                          ('player_info', 3, 'M:PlayerInfo', 'rep'),
                          ('start_raw', 4, 'M:StartRaw'),
                          ('options', 5, 'M:InterfaceOptions'),
                          ('mod_names', 6, 'str', 'rep')]),
    ('RequestObservation', [('disable_fog', 1, 'bool'),
                            ('game_loop', 2, 'u32')]),
    ('PlayerCommon', [('player_id', 1, 'u32'), ('minerals', 2, 'u32'),
                      ('vespene', 3, 'u32'), ('food_cap', 4, 'u32'),
                      ('food_used', 5, 'u32'), ('food_army', 6, 'u32'),
                      ('food_workers', 7, 'u32'),
                      ('idle_worker_count', 8, 'u32'),
                      ('army_count', 9, 'u32'), ('warp_gate_count', 10, 'u32'),
                      ('larva_count', 11, 'u32')]),
    ('AvailableAbility', [('ability_id', 1, 'i32'),
                          ('requires_point', 2, 'bool')]),
    ('Observation', [('player_common', 1, 'M:PlayerCommon'),
                     ('abilities', 3, 'M:AvailableAbility', 'rep'),
                     ('score', 4, 'M:Score'),
                     ('raw_data', 5, 'M:ObservationRaw'),
                     ('feature_layer_data', 6, 'M:ObservationFeatureLayer'),
                     ('game_loop', 9, 'u32'),
                     ('alerts', 10, 'i32', 'rep')]),
    ('ActionChat', [('channel', 1, 'i32'), ('message', 2, 'str')]),
    ('Action', [('action_raw', 1, 'M:ActionRaw'),
                ('action_chat', 6, 'M:ActionChat'),
                ('game_loop', 7, 'u32')]),
    ('ActionError', [('unit_tag', 1, 'u64'), ('ability_id', 2, 'u64'),
                     ('result', 3, 'i32')]),
    ('PlayerResult', [('player_id', 1, 'u32'), ('result', 2, 'i32')]),
    ('ChatReceived', [('player_id', 1, 'u32'), ('message', 2, 'str')]),
    ('ResponseObservation', [('actions', 1, 'M:Action', 'rep'),
                             ('action_errors', 2, 'M:ActionError', 'rep'),
                             ('observation', 3, 'M:Observation'),
                             ('player_result', 4, 'M:PlayerResult', 'rep'),
                             ('chat', 5, 'M:ChatReceived', 'rep')]),
    ('RequestAction', [('actions', 1, 'M:Action', 'rep')]),
    ('ResponseAction', [('result', 1, 'i32', 'rep')]),
    ('RequestStep', [('count', 1, 'u32')]),
    ('ResponseStep', [('simulation_loop', 1, 'u32')]),
    ('RequestData', [('ability_id', 1, 'bool'), ('unit_type_id', 2, 'bool'),
                     ('upgrade_id', 3, 'bool'), ('buff_id', 4, 'bool'),
                     ('effect_id', 5, 'bool')]),
    ('ResponseData', [('abilities', 1, 'M:AbilityData', 'rep'),
                      ('units', 2, 'M:UnitTypeData', 'rep'),
                      ('upgrades', 3, 'M:UpgradeData', 'rep'),
                      ('buffs', 4, 'M:BuffData', 'rep'),
                      ('effects', 5, 'M:EffectData', 'rep')]),
    ('RequestQuery', []),
    ('ResponseQuery', []),
    ('RequestSaveReplay', []),
    ('ResponseSaveReplay', [('data', 1, 'bytes')]),
    ('PlayerInfoExtra', [('player_info', 1, 'M:PlayerInfo'),
                         ('player_result', 2, 'M:PlayerResult'),
                         ('player_mmr', 3, 'i32'), ('player_apm', 4, 'i32')]),
    ('RequestReplayInfo', [('replay_path', 1, 'str'),
                           ('replay_data', 2, 'bytes'),
                           ('download_data', 3, 'bool')]),
    ('ResponseReplayInfo', [('map_name', 1, 'str'),
                            ('local_map_path', 2, 'str'),
                            ('player_info', 3, 'M:PlayerInfoExtra', 'rep'),
                            ('game_duration_loops', 4, 'u32'),
                            ('game_duration_seconds', 5, 'f32'),
                            ('game_version', 6, 'str'),
                            ('data_build', 7, 'u32'),
                            ('base_build', 8, 'u32'),
                            ('data_version', 11, 'str')]),
    ('RequestAvailableMaps', []),
    ('ResponseAvailableMaps', [('local_map_paths', 1, 'str', 'rep'),
                               ('battlenet_map_names', 2, 'str', 'rep')]),
    ('RequestSaveMap', [('map_path', 1, 'str'), ('map_data', 2, 'bytes')]),
    ('ResponseSaveMap', [('error', 1, 'i32')]),
    ('RequestPing', []),
    ('ResponsePing', [('game_version', 1, 'str'), ('data_version', 2, 'str'),
                      ('data_build', 3, 'u32'), ('base_build', 4, 'u32')]),
    ('RequestDebug', []),     # DebugCommand payloads need the official pkg
    ('ResponseDebug', []),
    ('Request', [('create_game', 1, 'M:RequestCreateGame'),
                 ('join_game', 2, 'M:RequestJoinGame'),
                 ('restart_game', 3, 'M:RequestRestartGame'),
                 ('start_replay', 4, 'M:RequestStartReplay'),
                 ('leave_game', 5, 'M:RequestLeaveGame'),
                 ('quit', 8, 'M:RequestQuit'),
                 ('game_info', 9, 'M:RequestGameInfo'),
                 ('observation', 10, 'M:RequestObservation'),
                 ('action', 11, 'M:RequestAction'),
                 ('step', 12, 'M:RequestStep'),
                 ('data', 13, 'M:RequestData'),
                 ('query', 14, 'M:RequestQuery'),
                 ('save_replay', 15, 'M:RequestSaveReplay'),
                 ('replay_info', 16, 'M:RequestReplayInfo'),
                 ('available_maps', 17, 'M:RequestAvailableMaps'),
                 ('save_map', 18, 'M:RequestSaveMap'),
                 ('ping', 19, 'M:RequestPing'),
                 ('debug', 20, 'M:RequestDebug'),
                 ('id', 97, 'u32')]),
    ('Response', [('create_game', 1, 'M:ResponseCreateGame'),
                  ('join_game', 2, 'M:ResponseJoinGame'),
                  ('restart_game', 3, 'M:ResponseRestartGame'),
                  ('start_replay', 4, 'M:ResponseStartReplay'),
                  ('leave_game', 5, 'M:ResponseLeaveGame'),
                  ('quit', 8, 'M:ResponseQuit'),
                  ('game_info', 9, 'M:ResponseGameInfo'),
                  ('observation', 10, 'M:ResponseObservation'),
                  ('action', 11, 'M:ResponseAction'),
                  ('step', 12, 'M:ResponseStep'),
                  ('data', 13, 'M:ResponseData'),
                  ('query', 14, 'M:ResponseQuery'),
                  ('save_replay', 15, 'M:ResponseSaveReplay'),
                  ('replay_info', 16, 'M:ResponseReplayInfo'),
                  ('available_maps', 17, 'M:ResponseAvailableMaps'),
                  ('save_map', 18, 'M:ResponseSaveMap'),
                  ('ping', 19, 'M:ResponsePing'),
                  ('debug', 20, 'M:ResponseDebug'),
                  ('id', 97, 'u32'),
                  ('error', 98, 'str', 'rep'),
                  ('status', 99, 'i32')]),
]

# Enum constants (values from the public protos).
_ENUMS = {
    # PlayerType
    'Participant': 1, 'Computer': 2, 'Observer': 3,
    # Race
    'NoRace': 0, 'Terran': 1, 'Zerg': 2, 'Protoss': 3, 'Random': 4,
    # Difficulty
    'VeryEasy': 1, 'Easy': 2, 'Medium': 3, 'MediumHard': 4, 'Hard': 5,
    'Harder': 6, 'VeryHard': 7, 'CheatVision': 8, 'CheatMoney': 9,
    'CheatInsane': 10,
    # AIBuild
    'RandomBuild': 1, 'Rush': 2, 'Timing': 3, 'Power': 4, 'Macro': 5,
    'Air': 6,
    # Result
    'Victory': 1, 'Defeat': 2, 'Tie': 3, 'Undecided': 4,
    # Status
    'launched': 1, 'init_game': 2, 'in_game': 3, 'in_replay': 4, 'ended': 5,
    'quit': 6, 'unknown': 99,
    # Alliance
    'Self': 1, 'Ally': 2, 'Neutral': 3, 'Enemy': 4,
    # DisplayType
    'Visible': 1, 'Snapshot': 2, 'Hidden': 3, 'Placeholder': 4,
    # ActionResult (success only; failure codes are game-data)
    'Success': 1,
    # ChatChannel
    'Broadcast': 1, 'Team': 2,
}

_LOCK = threading.Lock()
_NAMESPACE = None


class _Namespace:
    """Attribute bag exposing the message classes + enum constants, shaped
    like the official `s2clientprotocol.sc2api_pb2` module for the fields
    this repo uses."""

    def __init__(self, classes):
        for name, cls in classes.items():
            setattr(self, name, cls)
        for name, val in _ENUMS.items():
            setattr(self, name, val)


def _build():
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = 'distar_sc2.proto'
    fdp.package = _PKG
    fdp.syntax = 'proto2'
    for msg_name, fields in _MESSAGES:
        m = fdp.message_type.add()
        m.name = msg_name
        for spec in fields:
            fname, number, ftype = spec[0], spec[1], spec[2]
            rep = len(spec) > 3 and spec[3] == 'rep'
            f = m.field.add()
            f.name = fname
            f.number = number
            f.label = _F.LABEL_REPEATED if rep else _F.LABEL_OPTIONAL
            if ftype.startswith('M:'):
                f.type = _F.TYPE_MESSAGE
                f.type_name = f'.{_PKG}.{ftype[2:]}'
            else:
                f.type = _TYPES[ftype]
    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)
    try:
        classes = message_factory.GetMessageClassesForFiles(
            ['distar_sc2.proto'], pool)
    except AttributeError:                       # protobuf < 4.22
        factory = message_factory.MessageFactory(pool)
        classes = factory.GetMessages(['distar_sc2.proto'])
    return _Namespace({k.split('.')[-1]: v for k, v in classes.items()})


def get_protos():
    """Return (namespace, source).  Prefers the official s2clientprotocol
    package (source='s2clientprotocol'); falls back to the bundled minimal
    definitions (source='bundled')."""
    global _NAMESPACE
    try:
        from s2clientprotocol import sc2api_pb2 as sc_pb    # noqa: F401
        from s2clientprotocol import common_pb2 as sc_common
        ns = _OfficialNamespace(sc_pb, sc_common)
        return ns, 's2clientprotocol'
    except ImportError:
        pass
    with _LOCK:
        if _NAMESPACE is None:
            _NAMESPACE = _build()
    return _NAMESPACE, 'bundled'


class _OfficialNamespace:
    """Flatten the official package's sc2api_pb2 (+ enum constants from
    common/sc2api) behind the same attribute surface as _Namespace."""

    def __init__(self, sc_pb, sc_common):
        self._modules = [sc_pb, sc_common]

    def __getattr__(self, name):
        for mod in self._modules:
            if hasattr(mod, name):
                return getattr(mod, name)
        raise AttributeError(name)
