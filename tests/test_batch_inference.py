"""Shared-slab GPU batch inference protocol (reference actor.py:268-299):
clients write obs + bump signals; the server ticks once all slots signalled."""
import threading

import pytest
import torch

from distar_amd.actor.batch_inference import (BatchInferenceServer,
                                              copy_input_data)
from distar_amd.lib.consts import fake_step_data
from distar_amd.models import Model
from distar_amd.utils.config import Config


@pytest.mark.timeout(600)
def test_batch_inference_slab_roundtrip():
    torch.manual_seed(0)
    model = Model(Config({'common': {'type': 'train'}}))
    env_num = 2
    server = BatchInferenceServer(model, env_num, device='cpu')

    def client(env_id):
        obs = fake_step_data(train=False, entity_num=48, randomize=True)
        obs['hidden_state'] = [(torch.zeros(384), torch.zeros(384))
                               for _ in range(3)]
        copy_input_data(server.shared_input, obs, data_idx=env_id)
        server.signals[env_id] += 1

    threads = [threading.Thread(target=client, args=(i,)) for i in range(env_num)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert server.serve_once(server.signals, server.shared_input,
                             server.shared_output, model.compute_logp_action)
    assert (server.signals == 0).all()
    out = server.shared_output
    assert out['action_info']['action_type'].shape == (env_num,)
    assert torch.isfinite(out['logit']['action_type']).all()
    assert (out['selected_units_num'] <= 64).all()


@pytest.mark.timeout(900)
def test_actor_batched_inference_mode():
    """Actor with gpu_batch_inference: agents route through the shared slab
    server (partial-batch ticks keep the serial in-process loop live)."""
    torch.manual_seed(0)
    from distar_amd.actor.actor import Actor
    cfg = Config({'actor': {'episode_num': 1, 'env_type': 'mock',
                            'gpu_batch_inference': True, 'use_cuda': False,
                            'traj_len': 4},
                  'env': {'player_num': 2, 'max_episode_steps': 4},
                  'common': {'experiment_name': 'test_actor_slab',
                             'type': 'train'}})
    actor = Actor(cfg)
    results = actor.run()
    actor._batch_server.stop()
    assert len(results) == 1
    assert all(agent._batch_server is not None for agent in actor._agents)


@pytest.mark.timeout(900)
def test_actor_batch_inference_multi_env():
    """env_num x players share one slab server: every (worker, player) slot
    gets its own signal lane (reference actor.py:268-299 with env_num>1)."""
    torch.manual_seed(0)
    from distar_amd.actor.actor import Actor
    cfg = Config({'actor': {'episode_num': 2, 'env_num': 2, 'env_type': 'mock',
                            'traj_len': 4, 'gpu_batch_inference': True,
                            'use_cuda': False},
                  'env': {'player_num': 2, 'max_episode_steps': 4},
                  'common': {'experiment_name': 'test_actor_slab_multi',
                             'type': 'train'}})
    actor = Actor(cfg)
    results = actor.run()
    assert len(results) >= 2
    assert actor._batch_server.env_num == 4      # 2 envs x 2 players
    actor._batch_server.stop()
