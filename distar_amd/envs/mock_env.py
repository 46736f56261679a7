"""Mock SC2 environment honoring the observation/action spec.

The reference keeps a no-binary test env (`pysc2/env/mock_sc2_env.py:28-60`)
so agent loops run without StarCraft II; this is our equivalent at the
*agent tensor interface*: reset/step return per-agent observation dicts of
exactly the `lib.consts` schemas, actions are the 6-head dicts the agent
emits, per-agent skip-step scheduling mirrors `envs/env.py:333-375`, and
episodes end after a configurable number of game loops with a win/loss
outcome.  Used by the actor's synthetic-rollout mode and the CPU tests.
"""
import random


from ..lib.consts import fake_step_data
from .map_info import get_map_size


class MockSC2Env:
    def __init__(self, cfg=None, agent_num=2, max_episode_steps=200,
                 entity_num_range=(32, 128), seed=None):
        env_cfg = (cfg or {}).get('env', {}) if cfg is not None else {}
        self._agent_num = env_cfg.get('player_num', agent_num)
        self._max_episode_steps = env_cfg.get('max_episode_steps', max_episode_steps)
        self._entity_range = entity_num_range
        self._rng = random.Random(seed)
        self._map_name = env_cfg.get('map_name', 'KingsCove')
        self.map_size = get_map_size(self._map_name)
        self._episode_steps = 0
        self._next_obs_step = [0] * self._agent_num
        self._episode_count = 0
        self.game_info = [{'map_name': self._map_name} for _ in range(self._agent_num)]

    def _make_obs(self, agent_idx):
        en = self._rng.randint(*self._entity_range)
        obs = fake_step_data(train=False, entity_num=en, randomize=True)
        obs['game_loop'] = self._episode_steps
        obs['map_name'] = self._map_name
        obs['action_result'] = [1]
        obs['raw_obs'] = None        # no protobuf in mock mode
        return obs

    def reset(self):
        self._episode_steps = 0
        self._next_obs_step = [0] * self._agent_num
        self._episode_count += 1
        return {i: self._make_obs(i) for i in range(self._agent_num)}

    def step(self, actions):
        """actions: {agent_idx: {'func_id', 'skip_steps', ...}} -> per-agent
        skip scheduling: the env advances min(next_obs_step) like the
        reference (`envs/env.py:333-375`), plus 0-3 random delay steps."""
        for idx, action in (actions or {}).items():
            skip = int(action.get('skip_steps', 0)) if isinstance(action, dict) else 0
            self._next_obs_step[idx] = self._episode_steps + skip + 1
        target = min(self._next_obs_step)
        delay = self._rng.choices([0, 1, 2, 3], weights=[1, 1, 1, 1])[0]
        self._episode_steps = target + delay
        done = self._episode_steps >= self._max_episode_steps
        obs = {}
        for i in range(self._agent_num):
            if self._next_obs_step[i] <= self._episode_steps or done:
                obs[i] = self._make_obs(i)
        if done:
            winner = self._rng.randint(0, self._agent_num - 1)
            rewards = {i: (1 if i == winner else -1) for i in range(self._agent_num)}
        else:
            rewards = {i: 0 for i in obs}
        infos = {i: {'episode_steps': self._episode_steps} for i in obs}
        return obs, rewards, done, infos

    def close(self):
        pass
