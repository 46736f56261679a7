"""Map name -> playable-area size lookup (reference `distar/envs/map_info.py`),
plus localized Battle.net aliases for the ladder maps this framework ships
configs for.  Sizes are (x, y) playable dimensions used to clamp/scale
locations into the fixed 160x152 spatial grid."""

MAP_SIZES = {
    'KairosJunction': (144, 152),
    'KingsCove': (144, 152),
    'NewRepugnancy': (148, 132),
    'CyberForest': (152, 136),
    'GrassAndFire': (162, 170),
    'Automaton': (148, 148),
    'PortAleksander': (144, 160),
    'YearZero': (144, 148),
    'Thunderbird': (148, 140),
    'Acropolis': (160, 148),
    'Triton': (172, 164),
    'WintersGate': (144, 148),
    'WorldofSleepers': (152, 160),
    'Ephemeron': (128, 136),
    'DiscoBloodbath': (148, 148),
}

# localized names seen in Battle.net replays -> canonical
MAP_ALIASES = {
    '凯罗斯废料场': 'KairosJunction',
    '国王藏宝地': 'KingsCove',
    '新生雷帕格斯': 'NewRepugnancy',
    'KairosJunctionLE': 'KairosJunction',
    "King's Cove LE": 'KingsCove',
    'New Repugnancy LE': 'NewRepugnancy',
}

DEFAULT_MAP_SIZE = (144, 152)


def get_map_size(map_name, padding=False):
    """-> (x, y); unknown maps fall back to the default ladder size."""
    name = MAP_ALIASES.get(map_name, map_name)
    name = name.replace(' ', '').replace('LE', '')
    size = MAP_SIZES.get(name, DEFAULT_MAP_SIZE)
    if padding:
        from ..lib.consts import SPATIAL_SIZE
        return (SPATIAL_SIZE[1], SPATIAL_SIZE[0])
    return size


LADDER_MAPS = list(MAP_SIZES.keys())[:8]
