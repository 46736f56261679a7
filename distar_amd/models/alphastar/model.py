"""Top-level AlphaStar model with its five forward modes
(reference `distar/agent/default/model/model.py`):

  - ``forward``               plain step (env interaction)
  - ``compute_logp_action``   actor inference (sampled actions + log-probs)
  - ``compute_teacher_logit`` teacher logits for the KL term
  - ``rl_learner_forward``    RL learner: (T+1)-unroll, 6 value heads
  - ``sl_train``              SL learner: T-unroll teacher forcing

Checkpoint layout (module names) matches the reference exactly so reference
checkpoints load drop-in (tests/test_model_parity.py checks every key/shape
against assets/ckpt_layout_golden.json).
"""
import os.path as osp
from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn
from torch import Tensor

from .encoders import Encoder, ValueEncoder
from .policy import Policy
from .value import ValueBaseline
from ..nn.lnlstm import script_lnlstm
from ...lib.consts import MAX_SELECTED_UNITS_NUM
from ...utils.config import read_config, deep_merge_dicts

alphastar_model_default_config = read_config(
    osp.join(osp.dirname(__file__), 'actor_critic_default_config.yaml'))


def detach_grad(data):
    if isinstance(data, dict):
        return {k: detach_grad(v) for k, v in data.items()}
    if isinstance(data, torch.Tensor):
        return data.detach()
    return data


class Model(nn.Module):
    def __init__(self, cfg=None, use_value_network=False, temperature=None):
        super().__init__()
        self.whole_cfg = deep_merge_dicts(alphastar_model_default_config, cfg or {})
        if temperature is not None:
            self.whole_cfg.model.temperature = temperature
        self.cfg = self.whole_cfg.model
        self.encoder = Encoder(self.whole_cfg)
        self.policy = Policy(self.whole_cfg)
        self._use_value_feature = self.whole_cfg.learner.get('use_value_feature', False)
        if use_value_network:
            if self._use_value_feature:
                self.value_encoder = ValueEncoder(self.whole_cfg)
            self.value_networks = nn.ModuleDict()
            for k, v in self.cfg.value.items():
                if k in self.cfg.enable_baselines:
                    self.value_networks[v.name] = ValueBaseline(v.param, self._use_value_feature)
        self.only_update_baseline = self.cfg.get('only_update_baseline', False)
        self.core_lstm = script_lnlstm(self.cfg.encoder.core_lstm.input_size,
                                       self.cfg.encoder.core_lstm.hidden_size,
                                       self.cfg.encoder.core_lstm.num_layers)

    # ------------------------------------------------------------- env step
    def forward(self, spatial_info: Dict[str, Tensor], entity_info: Dict[str, Tensor],
                scalar_info: Dict[str, Tensor], entity_num: Tensor,
                hidden_state: List[Tuple[Tensor, Tensor]]):
        lstm_input, scalar_context, baseline_feature, entity_embeddings, map_skip = \
            self.encoder(spatial_info, entity_info, scalar_info, entity_num)
        lstm_output, out_state = self.core_lstm(lstm_input.unsqueeze(0), hidden_state)
        action_info, selected_units_num, logit, extra_units = self.policy(
            lstm_output.squeeze(0), entity_embeddings, map_skip, scalar_context, entity_num)
        return action_info, selected_units_num, out_state

    # ------------------------------------------------------ actor inference
    def compute_logp_action(self, spatial_info, entity_info, scalar_info,
                            entity_num, hidden_state, **kwargs):
        lstm_input, scalar_context, baseline_feature, entity_embeddings, map_skip = \
            self.encoder(spatial_info, entity_info, scalar_info, entity_num)
        lstm_output, out_state = self.core_lstm(lstm_input.unsqueeze(0), hidden_state)
        action_info, selected_units_num, logit, extra_units = self.policy(
            lstm_output.squeeze(0), entity_embeddings, map_skip, scalar_context, entity_num)
        log_action_probs = {}
        for k, action in action_info.items():
            dist = torch.distributions.Categorical(logits=logit[k])
            log_action_probs[k] = dist.log_prob(action)
        return {'action_info': action_info, 'action_logp': log_action_probs,
                'selected_units_num': selected_units_num, 'entity_num': entity_num,
                'hidden_state': out_state, 'logit': logit, 'extra_units': extra_units}

    # ------------------------------------------------------- teacher logits
    def compute_teacher_logit(self, spatial_info, entity_info, scalar_info,
                              entity_num, hidden_state, selected_units_num,
                              action_info, **kwargs):
        lstm_input, scalar_context, baseline_feature, entity_embeddings, map_skip = \
            self.encoder(spatial_info, entity_info, scalar_info, entity_num)
        lstm_output, out_state = self.core_lstm(lstm_input.unsqueeze(0), hidden_state)
        action_info, selected_units_num, logit = self.policy.train_forward(
            lstm_output.squeeze(0), entity_embeddings, map_skip, scalar_context,
            entity_num, action_info, selected_units_num)
        return {'logit': logit, 'hidden_state': out_state, 'entity_num': entity_num,
                'selected_units_num': selected_units_num}

    # ------------------------------------------------------------ RL learner
    def rl_learner_forward(self, spatial_info, entity_info, scalar_info, entity_num,
                           hidden_state, action_info, selected_units_num,
                           behaviour_logp, teacher_logit, mask, reward, step,
                           batch_size, unroll_len, **kwargs):
        flat_action_info = {k: torch.flatten(v, 0, 1) for k, v in action_info.items()}
        flat_selected_units_num = torch.flatten(selected_units_num, 0, 1)

        lstm_input, scalar_context, baseline_feature, entity_embeddings, map_skip = \
            self.encoder(spatial_info, entity_info, scalar_info, entity_num)
        hidden_size = hidden_state[0][0].shape[-1]
        # keep only the first frame's state per lane: (T+1)*B rows -> B
        hidden_state = [
            [hidden_state[i][j].view(-1, batch_size, hidden_size)[0] for j in range(2)]
            for i in range(len(hidden_state))
        ]
        lstm_output, out_state = self.core_lstm(
            lstm_input.view(-1, batch_size, lstm_input.shape[-1]), hidden_state)
        lstm_output = lstm_output.view(-1, lstm_output.shape[-1])

        policy_lstm_input = lstm_output[:-batch_size]
        policy_entity_embeddings = entity_embeddings[:-batch_size]
        policy_map_skip = [m[:-batch_size] for m in map_skip]
        policy_scalar_context = scalar_context[:-batch_size]
        policy_entity_num = entity_num[:-batch_size]
        _, _, logits = self.policy.train_forward(
            policy_lstm_input, policy_entity_embeddings, policy_map_skip,
            policy_scalar_context, policy_entity_num, flat_action_info,
            flat_selected_units_num)

        critic_input = lstm_output
        if self.only_update_baseline:
            critic_input = detach_grad(critic_input)
            baseline_feature = detach_grad(baseline_feature)
        if self._use_value_feature:
            value_feature = self.value_encoder(kwargs['value_feature'])
            critic_input = torch.cat([critic_input, value_feature, baseline_feature], dim=1)
        baseline_values = {k: v(critic_input) for k, v in self.value_networks.items()}

        logits = {k: v.view(unroll_len, batch_size, *v.shape[1:]) for k, v in logits.items()}
        baseline_values = {k: v.view(unroll_len + 1, batch_size)
                           for k, v in baseline_values.items()}
        logits['selected_units'] = torch.nn.functional.pad(
            logits['selected_units'],
            (0, 0, 0, MAX_SELECTED_UNITS_NUM - logits['selected_units'].shape[2]),
            'constant', -1e9)
        return {
            'unroll_len': unroll_len, 'batch_size': batch_size,
            'selected_units_num': selected_units_num, 'target_logit': logits,
            'value': baseline_values, 'action_log_prob': behaviour_logp,
            'teacher_logit': teacher_logit, 'mask': mask, 'action': action_info,
            'reward': reward, 'step': step,
        }

    # ------------------------------------------------------------ SL learner
    def sl_train(self, spatial_info, entity_info, scalar_info, entity_num,
                 selected_units_num, traj_lens, hidden_state, action_info, **kwargs):
        batch_size = len(traj_lens)
        lstm_input, scalar_context, baseline_feature, entity_embeddings, map_skip = \
            self.encoder(spatial_info, entity_info, scalar_info, entity_num)
        # rows are (B, T) lane-major from the SL dataloader: reshape + permute
        lstm_input = lstm_input.view(-1, lstm_input.shape[0] // batch_size,
                                     lstm_input.shape[-1]).permute(1, 0, 2)
        lstm_output, out_state = self.core_lstm(lstm_input, hidden_state)
        lstm_output = lstm_output.permute(1, 0, 2).contiguous().view(-1, lstm_output.shape[-1])
        action_info, selected_units_num, logits = self.policy.train_forward(
            lstm_output, entity_embeddings, map_skip, scalar_context, entity_num,
            action_info, selected_units_num)
        return logits, action_info, out_state
