"""Descriptor-generated observation builders: the same synthetic content as
tests/dummy_obs.py but as REAL protobuf messages (distar_amd.lib.sc2_protos),
following the reference's `pysc2/lib/features_test.py` pattern of testing
obs transforms against genuine `ResponseObservation` protos.  Field-name
drift between lib/features.py and the proto schema fails here, where the
duck-typed builders could silently mask it.
"""
import numpy as np

from distar_amd.lib.sc2_protos import get_protos

pb, PROTO_SOURCE = get_protos()


def image(msg, data, bpp=8):
    data = np.asarray(data, dtype=np.uint8)
    msg.bits_per_pixel = bpp
    msg.size.y, msg.size.x = data.shape
    msg.data = data.tobytes()


def add_unit(raw, tag, unit_type=86, alliance=1, x=30.0, y=30.0, orders=(),
             buffs=(), health=100, health_max=100, passengers=()):
    u = raw.units.add()
    u.tag = tag
    u.unit_type = unit_type
    u.alliance = alliance
    u.display_type = 1
    u.owner = 1 if alliance == 1 else 2
    u.pos.x, u.pos.y = x, y
    u.cloak = 3
    u.is_powered = True
    u.is_active = True
    u.build_progress = 1.0
    u.health, u.health_max = health, health_max
    u.shield = u.shield_max = 0
    u.energy = u.energy_max = 0
    for ab in orders:
        o = u.orders.add()
        o.ability_id = ab
        o.progress = 0.5
    u.buff_ids.extend(buffs)
    for ptag in passengers:
        p = u.passengers.add()
        p.tag = ptag
        p.unit_type = 105
        p.health = p.health_max = 35
    return u


def response_observation(num_units=8, game_loop=100, map_y=152, map_x=160,
                         player_id=1, upgrades=(), with_result=None,
                         actions=()):
    resp = pb.ResponseObservation()
    ob = resp.observation
    ob.game_loop = game_loop
    pc = ob.player_common
    pc.player_id = player_id
    pc.minerals, pc.vespene = 50, 0
    pc.food_used, pc.food_cap = 12, 14
    pc.food_army, pc.food_workers = 0, 12
    pc.idle_worker_count = 0
    pc.army_count, pc.warp_gate_count, pc.larva_count = 0, 0, 3
    ob.raw_data.player.upgrade_ids.extend(upgrades)
    # drones + a hatchery around the start location
    add_unit(ob.raw_data, tag=1000, unit_type=86, x=30.0, y=30.0)   # hatchery
    for i in range(num_units - 1):
        add_unit(ob.raw_data, tag=2000 + i, unit_type=104,
                 x=28.0 + i % 5, y=32.0 + i // 5)
    sd = ob.score.score_details
    for cat in ('none', 'army', 'economy', 'technology', 'upgrade'):
        setattr(sd.killed_minerals, cat, 100.0 if cat == 'army' else 0.0)
        setattr(sd.killed_vespene, cat, 50.0 if cat == 'army' else 0.0)
    h = np.zeros((map_y, map_x), dtype=np.uint8)
    mm = ob.feature_layer_data.minimap_renders
    for name in ('height_map', 'visibility_map', 'creep', 'player_relative',
                 'alerts', 'pathable', 'buildable'):
        image(getattr(mm, name), h)
    if with_result is not None:
        for pid, res in enumerate(with_result, start=1):
            pr = resp.player_result.add()
            pr.player_id = pid
            pr.result = res
    for act in actions:
        resp.actions.append(act)
    return resp


def game_info(map_y=152, map_x=160, map_name='KingsCove'):
    gi = pb.ResponseGameInfo()
    gi.map_name = map_name
    gi.start_raw.map_size.x, gi.start_raw.map_size.y = map_x, map_y
    loc = gi.start_raw.start_locations.add()
    loc.x, loc.y = 120.0, 120.0
    for pid in (1, 2):
        pi = gi.player_info.add()
        pi.player_id = pid
        pi.type = pb.Participant
        pi.race_requested = pb.Zerg
    return gi


def raw_action(ability_id, unit_tags=(), target_unit_tag=None,
               target_pos=None, queue_command=False, game_loop=0):
    act = pb.Action()
    uc = act.action_raw.unit_command
    uc.ability_id = ability_id
    uc.unit_tags.extend(unit_tags)
    uc.queue_command = queue_command
    if target_unit_tag is not None:
        uc.target_unit_tag = target_unit_tag
    if target_pos is not None:
        uc.target_world_space_pos.x = target_pos[0]
        uc.target_world_space_pos.y = target_pos[1]
    if game_loop:
        act.game_loop = game_loop
    return act
