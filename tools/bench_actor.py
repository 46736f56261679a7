#!/usr/bin/env python3
"""Actor inference latency bench (reference anchor: 0.16 s per 16-env GPU
batch, docs/guidance_to_small_scale_training.md:232).  Runs
Model.compute_logp_action on a 16-env synthetic batch with and without the
HIP kernels and prints ms per batched inference."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from distar_amd.lib.fake_data import fake_obs_step                 # noqa: E402
from distar_amd.models import Model                                # noqa: E402
from distar_amd.utils.config import Config                         # noqa: E402
from distar_amd.utils.data import default_collate_with_dim, to_device  # noqa: E402


def main(envs=16, entity_num=256, iters=10):
    device = 'cuda' if torch.cuda.is_available() else 'cpu'
    torch.manual_seed(0)
    model = Model(Config({'common': {'type': 'train'}})).to(device)
    obs = to_device(default_collate_with_dim(
        [fake_obs_step(entity_num=entity_num) for _ in range(envs)]), device)
    hidden = [(torch.zeros(envs, 384, device=device),
               torch.zeros(envs, 384, device=device)) for _ in range(3)]
    modes = [('hip', '0'), ('eager', '1')] if device == 'cuda' else [('cpu', '1')]
    for tag, flag in modes:
        os.environ['DISTAR_AMD_DISABLE_HIP'] = flag
        with torch.no_grad():
            for _ in range(3):
                model.compute_logp_action(**obs, hidden_state=hidden)
            if device == 'cuda':
                torch.cuda.synchronize()
            t0 = time.time()
            for _ in range(iters):
                model.compute_logp_action(**obs, hidden_state=hidden)
            if device == 'cuda':
                torch.cuda.synchronize()
        ms = (time.time() - t0) / iters * 1000
        print(f'{tag}: {ms:.2f} ms per {envs}-env batched inference '
              f'({ms / envs:.2f} ms/env)')


if __name__ == '__main__':
    main()
