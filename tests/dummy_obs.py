"""Synthetic raw-observation builders shaped like s2clientprotocol messages
(the pysc2 `dummy_observation` pattern): duck-typed namespaces
so `lib.features.Features` runs with no protobuf dependency."""
from types import SimpleNamespace as NS

import numpy as np


def image(data, bpp=8):
    data = np.asarray(data, dtype=np.uint8)
    return NS(bits_per_pixel=bpp, size=NS(y=data.shape[0], x=data.shape[1]),
              data=data.tobytes())


def passenger(tag, unit_type=105, health=35, health_max=35):
    return NS(tag=tag, unit_type=unit_type, health=health, health_max=health_max,
              shield=0, shield_max=0, energy=0, energy_max=0)


def unit(tag, unit_type=86, alliance=1, x=30.0, y=30.0, orders=(), buffs=(),
         health=100, health_max=100, passengers=()):
    return NS(tag=tag, unit_type=unit_type, alliance=alliance,
              cargo_space_taken=0, build_progress=1.0, health_max=health_max,
              shield_max=0, energy_max=0, display_type=1, owner=1,
              pos=NS(x=x, y=y), cloak=3, is_blip=False, is_powered=True,
              mineral_contents=0, vespene_contents=0, cargo_space_max=0,
              assigned_harvesters=0, weapon_cooldown=0,
              orders=[NS(ability_id=a, progress=0.5) for a in orders],
              is_hallucination=False, buff_ids=list(buffs), add_on_tag=0,
              is_active=True, attack_upgrade_level=0, armor_upgrade_level=0,
              shield_upgrade_level=0, health=health, shield=0, energy=0,
              passengers=list(passengers))


def score():
    cat = NS(none=0., army=100., economy=50., technology=0., upgrade=0.)
    return NS(score_details=NS(killed_minerals=cat, killed_vespene=cat))


def raw_observation(units, game_loop=100, map_y=152, map_x=160, player_id=1,
                    upgrades=(), effects=()):
    h = np.zeros((map_y, map_x), dtype=np.uint8)
    mm = NS(height_map=image(h), visibility_map=image(h), creep=image(h),
            player_relative=image(h), alerts=image(h), pathable=image(h),
            buildable=image(h))
    return NS(observation=NS(
        game_loop=game_loop,
        raw_data=NS(units=units,
                    effects=[NS(effect_id=e[0], owner=e[1],
                                pos=[NS(x=e[2], y=e[3])]) for e in effects],
                    player=NS(upgrade_ids=list(upgrades))),
        player_common=NS(player_id=player_id, minerals=50, vespene=0,
                         food_used=12, food_cap=14, food_army=0,
                         food_workers=12, idle_worker_count=0, army_count=0,
                         warp_gate_count=0, larva_count=3),
        feature_layer_data=NS(minimap_renders=mm),
        score=score()),
        action_errors=[], player_result=[])


def game_info(map_y=152, map_x=160, map_name='KingsCove'):
    return NS(start_raw=NS(map_size=NS(x=map_x, y=map_y),
                           start_locations=[NS(x=120.0, y=120.0)]),
              player_info=[NS(player_id=1, race_requested=2, type=1),
                           NS(player_id=2, race_requested=2, type=1)],
              map_name=map_name)


def raw_action(ability_id=None, unit_tags=(), target_unit_tag=None,
               target_pos=None, queue_command=False):
    uc = NS(ability_id=ability_id, unit_tags=list(unit_tags),
            queue_command=queue_command)
    fields = set()
    if target_unit_tag is not None:
        uc.target_unit_tag = target_unit_tag
        fields.add('target_unit_tag')
    if target_pos is not None:
        uc.target_world_space_pos = NS(x=target_pos[0], y=target_pos[1])
        fields.add('target_world_space_pos')
    uc.HasField = lambda f, fields=fields: f in fields
    top_fields = {'unit_command'} if ability_id is not None else set()
    act = NS(unit_command=uc)
    act.HasField = lambda f, tf=top_fields: f in tf
    return NS(action_raw=act)
