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
