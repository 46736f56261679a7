"""Template third-party agent (reference `distar/agent/template/agent.py` +
docs/agent.md): the minimal surface a custom pipeline must provide."""
import torch

from ...lib.consts import MAX_DELAY


class Agent:
    HAS_MODEL = False
    HAS_TEACHER = False

    def __init__(self, cfg=None, env_id=0):
        self.player_id = 'template'
        self.race = 'zerg'

    def reset(self, map_name='KingsCove', race='zerg', opponent_race='zerg',
              obs=None):
        self.race = race

    def step(self, observation):
        """Return one no-op action; replace with real inference."""
        return [{'func_id': 0, 'skip_steps': int(torch.randint(0, MAX_DELAY, ())),
                 'queued': 0, 'unit_tags': [], 'target_unit_tag': 0,
                 'location': (0, 0)}]

    def collect_data(self, next_obs, reward, done, idx):
        return None

    def get_unit_num_info(self):
        return {'unit_num': {}}

    def get_stat_data(self):
        return {'race_id': self.race, 'z_type': 0}
