"""Shared-memory GPU batch inference for actor fleets.

Functional parity with the reference's optional batched-inference mode
(`actor/actor.py:268-299`, `agent/default/agent.py:128-141,298-319,715-739`):
env processes write observations into a pre-allocated shared-memory input
slab at their slot and bump a per-slot signal counter; a single GPU server
loop collates nothing (the slab IS the batch), runs
`Model.compute_logp_action` (and a teacher `compute_teacher_logit` slab),
writes results into the shared output slab, and zeroes the signals.

MI355X notes: one H2D copy of the whole slab per tick (no per-env copies),
inference under bf16 autocast, outputs staged back through a pinned buffer.
Signals are shared int tensors — plain shared-memory polling like the
reference (works across fork/spawn since tensors are allocated in shared
memory before workers start).
"""
import time

import torch

from ..lib.consts import fake_model_output, fake_step_data
from ..utils.data import to_device


def copy_input_data(shared_input, step_data, data_idx):
    """Write one env's obs (+hidden) into slab slot ``data_idx`` (reference
    agent.py:37-89)."""
    for k, v in step_data.items():
        if k == 'hidden_state':
            for layer, (h, c) in enumerate(v):
                shared_input['hidden_state'][layer][0][data_idx].copy_(h)
                shared_input['hidden_state'][layer][1][data_idx].copy_(c)
        elif isinstance(v, torch.Tensor):
            shared_input[k][data_idx].copy_(v)
        elif isinstance(v, dict):
            for _k, _v in v.items():
                if _k in shared_input[k]:
                    dst = shared_input[k][_k][data_idx]
                    if _v.shape == dst.shape:
                        dst.copy_(_v)
                    else:               # entity axis narrower than the slab
                        dst.zero_()
                        dst[tuple(slice(0, s) for s in _v.shape)].copy_(_v)


def copy_output_data(shared_output, model_output, batch_size):
    for k, v in model_output.items():
        if k == 'hidden_state':
            for layer, (h, c) in enumerate(v):
                shared_output['hidden_state'][layer][0].copy_(h.cpu())
                shared_output['hidden_state'][layer][1].copy_(c.cpu())
        elif isinstance(v, torch.Tensor):
            shared_output[k][:batch_size].copy_(v.cpu())
        elif isinstance(v, dict):
            for _k, _v in v.items():
                if _k in shared_output[k]:
                    dst = shared_output[k][_k]
                    src = _v.cpu()
                    if src.shape == dst.shape:
                        dst.copy_(src)
                    else:
                        dst.zero_()
                        dst[tuple(slice(0, s) for s in src.shape)].copy_(src)


class BatchInferenceServer:
    """One GPU process serving ``env_num`` rollout processes."""

    def __init__(self, model, env_num, hidden_size=384, hidden_layer=3,
                 device='cuda', teacher_model=None):
        self.model = model
        self.teacher_model = teacher_model
        self.env_num = env_num
        self.device = device
        self.shared_input = fake_step_data(share_memory=True, batch_size=env_num,
                                           train=False, hidden_size=hidden_size,
                                           hidden_layer=hidden_layer)
        self.shared_output = fake_model_output(env_num, hidden_size, hidden_layer)
        self.signals = torch.zeros(env_num, dtype=torch.long).share_memory_()
        if teacher_model is not None:
            self.teacher_input = fake_step_data(share_memory=True,
                                                batch_size=env_num, train=True,
                                                hidden_size=hidden_size,
                                                hidden_layer=hidden_layer)
            self.teacher_output = fake_model_output(env_num, hidden_size,
                                                    hidden_layer, teacher=True)
            self.teacher_signals = torch.zeros(env_num, dtype=torch.long).share_memory_()
        self._stop = False

    def serve_once(self, signals, shared_input, shared_output, forward,
                   force=False):
        """One tick when every slot has signalled — or, with ``force``, when
        at least one has (partial batches keep a serial/in-process actor
        from deadlocking; un-signalled slots' outputs are recomputed stale
        values nobody reads)."""
        pending = signals > 0
        if not bool(pending.any()):
            return False
        if not force and int(pending.sum()) < self.env_num:
            return False
        ticked = pending.clone()
        batch = {k: v for k, v in shared_input.items() if k != 'hidden_state'}
        batch = to_device(batch, self.device)
        batch['hidden_state'] = [
            (shared_input['hidden_state'][l][0].to(self.device),
             shared_input['hidden_state'][l][1].to(self.device))
            for l in range(len(shared_input['hidden_state']))]
        use_amp = str(self.device).startswith('cuda')
        with torch.no_grad(), torch.autocast('cuda', dtype=torch.bfloat16,
                                             enabled=use_amp):
            output = forward(**batch)
        copy_output_data(shared_output, output, self.env_num)
        signals[ticked] = 0
        return True

    def run(self, poll_interval=0.002, partial_after=0.01):
        pending_since = None
        while not self._stop:
            any_pending = bool((self.signals > 0).any())
            force = False
            if any_pending:
                if pending_since is None:
                    pending_since = time.time()
                force = time.time() - pending_since > partial_after
            ticked = self.serve_once(self.signals, self.shared_input,
                                     self.shared_output,
                                     self.model.compute_logp_action,
                                     force=force)
            if ticked:
                pending_since = None
            if self.teacher_model is not None:
                ticked |= self.serve_once(self.teacher_signals,
                                          self.teacher_input,
                                          self.teacher_output,
                                          self.teacher_model.compute_teacher_logit,
                                          force=force)
            if not ticked:
                time.sleep(poll_interval)

    def stop(self):
        self._stop = True
