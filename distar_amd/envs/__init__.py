from .mock_env import MockSC2Env
from .map_info import get_map_size, LADDER_MAPS
