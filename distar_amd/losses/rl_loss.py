"""Reinforcement loss: per-baseline V-trace PG + UPGO + TD(lambda) critics +
normalized entropy + teacher KL (+ optional DAPO successive-model KL).

Functional parity with the reference's `rl_training/rl_loss.py:9-199`.
"""
import os.path as osp

import torch

from .rl_utils import (dapo_loss, entropy_loss, kl_loss, policy_gradient_loss,
                       td_lambda_loss, upgo_loss)
from ..utils.config import deep_merge_dicts, read_config

default_config = read_config(osp.join(osp.dirname(__file__), 'default_reinforcement_loss.yaml'))


class ReinforcementLoss:
    def __init__(self, learner_cfg, player_id='MP0'):
        self.cfg = deep_merge_dicts(default_config.learner, learner_cfg)
        self.gammas = self.cfg.gammas
        self.loss_weights = self.cfg.loss_weights
        self.action_type_kl_steps = self.cfg.kl.action_type_kl_steps
        self.dapo_steps = self.cfg.dapo.dapo_steps
        self.use_dapo = self.cfg.use_dapo
        if 'MP' not in player_id:
            self.use_dapo = False
            self.loss_weights.dapo = 0.0
        self.dapo_head_weights = self.cfg.dapo_head_weights
        self.pg_head_weights = self.cfg.pg_head_weights
        self.upgo_head_weights = self.cfg.upgo_head_weights
        self.entropy_head_weights = self.cfg.entropy_head_weights
        self.kl_head_weights = self.cfg.kl_head_weights
        self.only_update_value = False
        self.use_total_rhos = self.cfg.get('use_total_rhos', False)

    def compute_loss(self, inputs):
        target_logits_dict = inputs['target_logit']        # (T, B, ...)
        baseline_values_dict = inputs['value']             # (T+1, B)
        behaviour_logp_dict = inputs['action_log_prob']    # (T, B)
        teacher_logits_dict = inputs['teacher_logit']      # (T, B, ...)
        masks_dict = inputs['mask']
        actions_dict = inputs['action']
        rewards_dict = inputs['reward']
        game_steps = inputs['step']
        # zero the bootstrap value when the final winloss reward is 0 (episode
        # not finished within this trajectory slice)
        flag = rewards_dict['winloss'][-1] == 0
        for field in baseline_values_dict:
            baseline_values_dict[field][-1] = baseline_values_dict[field][-1] * flag

        loss_info_dict = {}
        # per-head row statistics via the fused rowwise kernels on GPU
        # (ops/{ce_loss,rl_rowwise}.py; eager composition on CPU):
        # action log-prob = -(lse - logit[a]), per-row entropy, per-row
        # KL(teacher||target) — the reference's Categorical-based prepare
        # stage materializes full (T,B,C) probs AND log-probs per head
        # (`rl_loss.py:63-90`); here neither is ever built.
        from ..ops.ce_loss import masked_cross_entropy
        from ..ops.rl_rowwise import rowwise_entropy, rowwise_kl
        target_action_log_probs_dict = {}
        entropy_rows_dict = {}
        kl_rows_dict = {}
        dapo_rows_dict = {}
        num_classes_dict = {}
        clipped_rhos_dict = {}
        for head_type in ['action_type', 'delay', 'queued', 'target_unit',
                          'selected_units', 'target_location']:
            target_logits = target_logits_dict[head_type]
            actions = actions_dict[head_type]
            num_classes_dict[head_type] = target_logits.shape[-1]
            target_action_log_probs = -masked_cross_entropy(target_logits,
                                                            actions)
            entropy_rows_dict[head_type] = rowwise_entropy(target_logits)
            kl_rows_dict[head_type] = rowwise_kl(
                teacher_logits_dict[head_type], target_logits)
            if self.use_dapo:
                dapo_rows_dict[head_type] = rowwise_kl(
                    inputs['successive_logit'][head_type], target_logits)
            behaviour_action_log_probs = behaviour_logp_dict[head_type]
            with torch.no_grad():
                log_rhos = target_action_log_probs - behaviour_action_log_probs
                if head_type == 'selected_units':
                    log_rhos = (log_rhos * masks_dict['selected_units_mask']).sum(dim=-1)
                clipped_rhos = torch.exp(log_rhos).clamp_(max=1)
            if head_type == 'selected_units':
                target_action_log_probs = target_action_log_probs.masked_fill(
                    ~masks_dict['selected_units_mask'], 0).sum(-1)
            target_action_log_probs_dict[head_type] = target_action_log_probs
            clipped_rhos_dict[head_type] = clipped_rhos

        # policy gradient (V-trace) per enabled baseline
        total_policy_gradient_loss = 0
        for field, baseline in baseline_values_dict.items():
            reward = rewards_dict[field]
            field_loss, field_info = policy_gradient_loss(
                baseline, reward, target_action_log_probs_dict, clipped_rhos_dict,
                masks_dict, head_weights_dict=self.pg_head_weights, gamma=1.0,
                field=field)
            total_policy_gradient_loss = total_policy_gradient_loss + \
                self.loss_weights.pg[field] * field_loss
            for k, v in field_info.items():
                loss_info_dict[field + '/' + k] = v

        # UPGO on winloss
        total_upgo_loss, upgo_info = upgo_loss(
            baseline_values_dict['winloss'], rewards_dict['winloss'],
            target_action_log_probs_dict, clipped_rhos_dict,
            masks_dict['actions_mask'], self.upgo_head_weights)
        total_upgo_loss = total_upgo_loss * self.loss_weights.upgo.winloss
        loss_info_dict.update(upgo_info)

        # TD(lambda) critics
        total_critic_loss = 0
        for field, baseline in baseline_values_dict.items():
            reward = rewards_dict[field]
            critic_loss = td_lambda_loss(baseline, reward, masks_dict,
                                         gamma=self.gammas.baseline[field], field=field)
            total_critic_loss = total_critic_loss + \
                self.loss_weights.baseline[field] * critic_loss
            loss_info_dict[field + '/td'] = critic_loss.detach()
            loss_info_dict[field + '/reward'] = reward.float().mean()
            loss_info_dict[field + '/value'] = baseline.mean().detach()
        if 'battle' in rewards_dict:
            loss_info_dict['battle/reward'] = rewards_dict['battle'].float().mean()

        # entropy
        total_entropy_loss, entropy_info = entropy_loss(
            entropy_rows_dict, num_classes_dict, masks_dict,
            head_weights_dict=self.entropy_head_weights)
        total_entropy_loss = total_entropy_loss * self.loss_weights.entropy
        loss_info_dict.update(entropy_info)

        # teacher KL
        total_kl_loss, action_type_kl_loss, kl_info = kl_loss(
            kl_rows_dict, masks_dict, game_steps,
            action_type_kl_steps=self.action_type_kl_steps,
            head_weights_dict=self.kl_head_weights)
        total_kl_loss = total_kl_loss * self.loss_weights.kl
        action_type_kl_loss = action_type_kl_loss * self.loss_weights.action_type_kl
        loss_info_dict.update(kl_info)

        # DAPO
        if self.use_dapo:
            total_dapo_loss, dapo_info = dapo_loss(
                dapo_rows_dict, masks_dict,
                game_steps, dapo_steps=self.dapo_steps,
                head_weights_dict=self.dapo_head_weights)
            total_dapo_loss = total_dapo_loss * self.loss_weights.dapo
            loss_info_dict.update(dapo_info)
        else:
            total_dapo_loss = 0.0

        if self.only_update_value:
            total_loss = total_critic_loss
        else:
            total_loss = (total_policy_gradient_loss + total_upgo_loss +
                          total_critic_loss + total_entropy_loss + total_kl_loss +
                          action_type_kl_loss + total_dapo_loss)
        # ONE host sync for all scalar metrics (the reference's per-metric
        # .item() calls stall the device queue ~40x per step)
        tensor_keys = [k for k, v in loss_info_dict.items()
                       if torch.is_tensor(v)]
        if tensor_keys:
            flat = torch.stack([loss_info_dict[k].detach().float().reshape(())
                                for k in tensor_keys]).cpu()
            for k, val in zip(tensor_keys, flat.tolist()):
                loss_info_dict[k] = val
        loss_info_dict['total_loss'] = total_loss
        return loss_info_dict

    def reset(self, learner_cfg):
        self.cfg = deep_merge_dicts(self.cfg, learner_cfg)
        self.gammas = self.cfg.gammas
        self.loss_weights = self.cfg.loss_weights
        self.action_type_kl_steps = self.cfg.kl.action_type_kl_steps
        self.pg_head_weights = self.cfg.pg_head_weights
        self.upgo_head_weights = self.cfg.upgo_head_weights
        self.entropy_head_weights = self.cfg.entropy_head_weights
        self.kl_head_weights = self.cfg.kl_head_weights
        self.only_update_value = False
