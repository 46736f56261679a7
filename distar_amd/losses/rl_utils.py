"""RL return/loss math: V-trace, UPGO, TD(lambda), entropy, teacher-KL, DAPO.

Functional parity with the reference's
`distar/agent/default/rl_training/as_rl_utils.py` (per-head loss wrappers and
the reverse scans).  The (T, B) reverse scans dispatch to the HIP scan kernel
on ROCm devices (`distar_amd/ops/scans.py`) — one kernel instead of a
T-iteration Python loop — and fall back to the eager loop on CPU.
"""
import torch
import torch.nn.functional as F

from ..ops.scans import lambda_return_scan, vtrace_scan

HEAD_TYPES = ['action_type', 'delay', 'queued', 'selected_units', 'target_unit',
              'target_location']


def multistep_forward_view(rewards, gammas, bootstrap_values, lambda_):
    """result[T-1] = r[T-1] + g[T-1]*v[T];
    result[t] = r[t] + g[t]*(l[t]*result[t+1] + (1-l[t])*v[t+1])."""
    return lambda_return_scan(rewards, gammas, bootstrap_values, lambda_)


def generalized_lambda_returns(rewards, gammas, bootstrap_values, lambda_):
    if not isinstance(gammas, torch.Tensor):
        gammas = gammas * torch.ones_like(rewards)
    if not isinstance(lambda_, torch.Tensor):
        lambda_ = lambda_ * torch.ones_like(rewards)
    return multistep_forward_view(rewards, gammas, bootstrap_values[1:], lambda_)


def td_lambda_loss(values, rewards, mask=None, gamma=1.0, lambda_=0.8, field=None):
    with torch.no_grad():
        returns = generalized_lambda_returns(rewards, gamma, values, lambda_)
    loss = 0.5 * torch.pow(returns - values[:-1], 2)
    if field in ('build_order', 'built_unit', 'effect'):
        loss = loss * mask[field + '_mask']
    return loss.mean()


def upgo_returns(rewards, bootstrap_values):
    """lambda_t = 1[r_{t+1} + V_{t+2} >= V_{t+1}], shifted left one step."""
    lambdas = (rewards + bootstrap_values[1:]) >= bootstrap_values[:-1]
    lambdas = torch.cat([lambdas[1:], torch.ones_like(lambdas[-1:])], dim=0)
    return generalized_lambda_returns(rewards, 1.0, bootstrap_values, lambdas.float())


def vtrace_advantages(clipped_rhos, clipped_cs, rewards, bootstrap_values,
                      clipped_pg_rhos=None, gammas=1.0, lambda_=0.8):
    if not isinstance(gammas, torch.Tensor):
        gammas = gammas * torch.ones_like(rewards)
    if not isinstance(lambda_, torch.Tensor):
        lambda_ = lambda_ * torch.ones_like(rewards)
    vtrace_val = vtrace_scan(clipped_rhos, clipped_cs, rewards, bootstrap_values,
                             gammas, lambda_)
    if clipped_pg_rhos is None:
        clipped_pg_rhos = clipped_rhos
    return clipped_pg_rhos * (rewards + gammas * vtrace_val[1:] - bootstrap_values[:-1])


def policy_gradient_loss(baseline_value, reward, target_action_log_probs_dict,
                         clipped_rhos_dict, mask, head_weights_dict, gamma=1.0,
                         field=None):
    """Separate V-trace PG loss per head (reference as_rl_utils.py:1-28)."""
    total = 0.
    info = {}
    for head_type in HEAD_TYPES:
        clipped_rhos = clipped_rhos_dict[head_type]
        log_probs = target_action_log_probs_dict[head_type]
        with torch.no_grad():
            advantages = vtrace_advantages(clipped_rhos, clipped_rhos, reward,
                                           baseline_value, gammas=gamma, lambda_=1.0)
        loss = -advantages * log_probs
        if head_type not in ('action_type', 'delay'):
            loss = loss * mask['actions_mask'][head_type]
        if field in ('build_order', 'built_unit', 'effect'):
            loss = loss * mask[field + '_mask']
        loss = loss.mean()
        total = total + loss * head_weights_dict[head_type]
        info[head_type] = loss.detach()
    info['total'] = total.detach()
    return total, info


def upgo_loss(baseline_value, reward, target_action_log_probs_dict,
              clipped_rhos_dict, mask, head_weights_dict):
    total = 0.
    info = {}
    for head_type in HEAD_TYPES:
        clipped_rhos = clipped_rhos_dict[head_type]
        log_probs = target_action_log_probs_dict[head_type]
        with torch.no_grad():
            advantages = clipped_rhos * (upgo_returns(reward, baseline_value)
                                         - baseline_value[:-1])
        loss = -advantages * log_probs
        if head_type not in ('action_type', 'delay'):
            loss = loss * mask[head_type]
        loss = loss.mean()
        total = total + loss * head_weights_dict[head_type]
        info['upgo/' + head_type] = loss.detach()
    info['upgo/total'] = total.detach()
    return total, info


def entropy_loss(entropy_rows_dict, num_classes_dict, mask, head_weights_dict):
    """Normalized entropy per head (reference as_rl_utils.py:52-75).

    `entropy_rows_dict[head]` is the per-row entropy -(p*logp).sum(-1)
    (rowwise_entropy — the fused HIP kernel on GPU, eager on CPU), so the
    (T,B,C) probs/log-probs tensors are never materialized here."""
    import math as _math
    total = 0.
    info = {}
    for head_type in HEAD_TYPES:
        ent = entropy_rows_dict[head_type]
        if head_type == 'selected_units':
            ent = ent / (1e-9 + torch.log(
                mask['selected_units_logits_mask'].float().sum(dim=-1) + 1).unsqueeze(-1))
            ent = (ent * mask['selected_units_mask']).sum(-1)
            ent = ent.div(mask['selected_units_mask'].sum(-1) + 1e-9)
        elif head_type == 'target_unit':
            ent = ent / (1e-9 + torch.log(
                mask['target_units_logits_mask'].float().sum(dim=-1) + 1))
        else:
            ent = ent / _math.log(num_classes_dict[head_type])
        if head_type not in ('action_type', 'delay'):
            ent = ent * mask['actions_mask'][head_type]
        entropy = ent.mean()
        info['entropy/' + head_type] = entropy.detach()
        total = total + (-entropy * head_weights_dict[head_type])
    info['entropy/total'] = total.detach()
    return total, info


def kl_loss(kl_rows_dict, mask, game_steps, action_type_kl_steps,
            head_weights_dict):
    """KL(teacher || target) per head, plus the early-game extra action-type
    KL (reference as_rl_utils.py:78-110).

    `kl_rows_dict[head]` is the per-row KL sum (rowwise_kl — fused on GPU),
    so teacher probs/log-probs are never materialized here."""
    total = 0.
    action_type_kl_loss = torch.tensor(0.)
    info = {}
    for head_type in ['action_type', 'queued', 'delay', 'selected_units',
                      'target_unit', 'target_location']:
        kl = kl_rows_dict[head_type]
        if head_type == 'selected_units':
            kl = (kl * mask['selected_units_mask']).sum(-1)
        if head_type not in ('action_type', 'delay'):
            kl = kl * mask['actions_mask'][head_type]
        if head_type == 'action_type':
            flag = game_steps < action_type_kl_steps
            action_type_kl = kl * flag * mask['cum_action_mask']
            action_type_kl_loss = action_type_kl.mean()
            info['kl/extra_at'] = action_type_kl_loss.detach()
        kl_head = kl.mean()
        total = total + kl_head * head_weights_dict[head_type]
        info['kl/' + head_type] = kl_head.detach()
    info['kl/total'] = total.detach()
    return total, action_type_kl_loss, info


def dapo_loss(dapo_rows_dict, mask, game_steps, dapo_steps,
              head_weights_dict):
    """KL against the successive model in the early game
    (reference as_rl_utils.py:112-136).  `dapo_rows_dict[head]` is the
    per-row KL(successive || target) sum."""
    total = 0.
    info = {}
    flag = game_steps < dapo_steps
    for head_type in HEAD_TYPES:
        kl = dapo_rows_dict[head_type]
        if head_type == 'selected_units':
            kl = (kl * mask['selected_units_mask']).sum(-1)
        if head_type not in ('action_type', 'delay'):
            kl = kl * mask['actions_mask'][head_type]
        kl = (kl * flag).mean()
        total = total + kl * head_weights_dict[head_type]
        info['dapo/' + head_type] = kl.detach()
    info['dapo/total'] = total.detach()
    return total, info
