"""Supervised (behaviour-cloning) loss over the six heads.

Functional parity with the reference's `sl_training/sl_loss.py:37-286`:
per-head masked CE + metrics, the selected-units label-availability mask
(`su_mask`), optional label smoothing, optional cross-rank renormalization by
the global batch (one scalar allreduce).
"""
import os.path as osp

import torch
import torch.nn.functional as F

from ..models.nn.blocks import sequence_mask
from ..ops.ce_loss import masked_cross_entropy
from ..parallel.dist import allreduce, get_rank, get_world_size
from ..utils.config import deep_merge_dicts, read_config

default_config = read_config(osp.join(osp.dirname(__file__), 'default_supervised_loss.yaml'))


class LabelSmoothingCrossEntropy(torch.nn.Module):
    def __init__(self, smoothing=0.1):
        super().__init__()
        assert smoothing < 1.0
        self.smoothing = smoothing
        self.confidence = 1. - smoothing

    def forward(self, x, target, reduce=False):
        logprobs = F.log_softmax(x, dim=-1)
        nll = -logprobs.gather(dim=-1, index=target.unsqueeze(1)).squeeze(1)
        smooth = -logprobs.mean(dim=-1)
        loss = self.confidence * nll + self.smoothing * smooth
        return loss.mean() if reduce else loss


class SupervisedLoss:
    def __init__(self, cfg):
        cfg = deep_merge_dicts(default_config, cfg)
        self.whole_cfg = cfg
        self.cfg = cfg.learner
        self.loss_func = {
            'action_type': self._action_type_loss,
            'delay': self._delay_loss,
            'queued': self._queued_loss,
            'selected_units': self._selected_units_loss,
            'target_unit': self._target_unit_loss,
            'target_location': self._target_location_loss,
        }
        self.loss_weight = self.cfg.loss_weight
        if self.cfg.get('label_smooth', False):
            self.criterion = LabelSmoothingCrossEntropy()
        else:
            self.criterion = torch.nn.CrossEntropyLoss(reduction='none')
        self.su_criterion = torch.nn.CrossEntropyLoss(reduction='none')
        self.label_smooth = self.cfg.get('label_smooth', False)
        self.su_mask = self.cfg.su_mask
        self.cross_rank_loss = self.cfg.get('cross_rank_loss', False)
        self.total_batch_size = None
        self.rank = get_rank()
        self.world_size = get_world_size()

    def compute_loss(self, policy_logits, actions, actions_mask, selected_units_num,
                     entity_num, infer_action_info):
        if self.cross_rank_loss:
            self.total_batch_size = torch.tensor(entity_num.shape[0], dtype=torch.float,
                                                 device=entity_num.device)
            allreduce(self.total_batch_size, average=False)
        loss_dict = {}
        for name, fn in self.loss_func.items():
            if name == 'selected_units':
                loss_dict.update(fn(policy_logits[name], actions[name], actions_mask[name],
                                    selected_units_num, entity_num,
                                    infer_action_info.get('selected_units')))
            else:
                loss_dict.update(fn(policy_logits[name], actions[name], actions_mask[name]))
        total = 0.
        for name in self.loss_func:
            total = total + loss_dict[name + '_loss'] * self.loss_weight[name]
        loss_dict['total_loss'] = total
        return loss_dict

    # ----------------------------------------------------------- helpers
    def _ce(self, logits, labels):
        """Per-row CE: the fused HIP kernel on GPU (K13 — one pass, no
        (N,C) log-softmax materialization), eager otherwise; label
        smoothing keeps the explicit composition."""
        if self.label_smooth:
            return self.criterion(logits, labels)
        return masked_cross_entropy(logits, labels)

    def _masked_mean(self, loss_tmp, mask, batch):
        if self.cross_rank_loss:
            loss = loss_tmp.mean()
            return (batch / self.total_batch_size * self.world_size) * loss
        valid = mask.sum()
        return loss_tmp.sum() / valid if valid > 0 else loss_tmp.sum() * 0

    def _action_type_loss(self, logits, labels, mask):
        with torch.no_grad():
            acc = (logits.argmax(dim=1) == labels).float().sum() / len(labels)
        loss_tmp = self._ce(logits, labels) * mask
        loss = self._masked_mean(loss_tmp, mask, labels.shape[0])
        return {'action_type_loss': loss, 'action_type_acc': acc}

    def _delay_loss(self, preds, labels, mask):
        loss_tmp = self._ce(preds, labels) * mask
        loss = self._masked_mean(loss_tmp, mask, labels.shape[0])
        with torch.no_grad():
            l1 = ((preds.argmax(dim=-1) - labels).abs() * mask).sum() / (mask.sum() + 1e-6)
        return {'delay_loss': loss, 'delay_distance_L1': l1}

    def _queued_loss(self, preds, labels, mask):
        loss_tmp = self._ce(preds, labels) * mask
        loss = self._masked_mean(loss_tmp, mask, labels.shape[0])
        with torch.no_grad():
            acc = ((preds.argmax(dim=-1) - labels).abs() * mask).sum() / (mask.sum() + 1e-6)
        return {'queued_loss': loss, 'queued_acc': acc}

    def _selected_units_loss(self, logits, labels, mask, lengths, entity_num, selected_units):
        b, s, n = logits.shape
        if self.su_mask:
            # restrict candidates to this sample's own labels (+ end flag):
            # zero out logits of units that are not in the label set
            length_wo_end = (lengths - 1).clamp(min=0)
            length_mask = sequence_mask(length_wo_end, max_len=labels.shape[1])
            new_labels = labels.clone()
            new_labels[~length_mask] = n          # park invalid labels in a scratch column
            new_labels = new_labels[:, :s]
            ext = torch.cat([logits, logits.new_zeros(b, s, 1)], dim=-1)
            logits_mask = torch.ones_like(ext)
            logits_mask = torch.scatter(logits_mask, 2,
                                        new_labels.unsqueeze(1).repeat(1, s, 1), 0.)
            logits_mask = torch.scatter(logits_mask, 2, new_labels.unsqueeze(2), 1.)
            ext = ext.masked_fill(~logits_mask.bool(), -1e9)
            logits = ext[:, :, :-1]
        select_mask = sequence_mask(lengths, max_len=s)
        loss_tmp = masked_cross_entropy(logits.reshape(-1, n),
                                        labels[:, :s].reshape(-1)).view(b, s)
        loss_tmp = loss_tmp.masked_fill(~select_mask, 0)
        loss_tmp = loss_tmp * mask.unsqueeze(1)
        if self.cross_rank_loss:
            loss = loss_tmp.sum() / b
            loss = (b / self.total_batch_size * self.world_size) * loss
        else:
            loss = loss_tmp.sum() / b
        loss_norm = loss_tmp.sum() / (lengths.sum() + 1e-6)
        end_flag_loss = loss_tmp[torch.arange(b, device=logits.device),
                                 (lengths - 1).clamp(min=0)].mean()
        with torch.no_grad():
            iou = self._selection_iou(selected_units, labels[:, :s], lengths, entity_num,
                                      mask, select_mask, s, n) \
                if selected_units is not None else torch.tensor(0.)
        return {'selected_units_loss': loss, 'selected_units_loss_norm': loss_norm,
                'selected_units_end_flag_loss': end_flag_loss, 'selected_units_iou': iou}

    @staticmethod
    def _selection_iou(preds, labels, lengths, entity_num, mask, select_mask, s, n):
        b = preds.shape[0]
        end_flag_index = (preds == entity_num.unsqueeze(1)).long()
        end_flag_index = torch.sort(end_flag_index, dim=-1, descending=True)[1][:, 0]
        invalid = end_flag_index == 0
        end_flag_index = end_flag_index + 1
        end_flag_index[invalid] += s
        preds = preds + 1
        labels = labels + 1
        preds_mask = sequence_mask(end_flag_index, max_len=preds.shape[1])
        labels = labels * select_mask
        preds = preds * preds_mask
        device = labels.device
        pred_set = torch.zeros(b, n + 2, dtype=torch.bool, device=device).scatter_(
            1, preds.long().clamp(max=n + 1), True)
        label_set = torch.zeros(b, n + 2, dtype=torch.bool, device=device).scatter_(
            1, labels.long().clamp(max=n + 1), True)
        inter = (pred_set & label_set)[:, 1:].sum(dim=1)
        union = (pred_set | label_set)[:, 1:].sum(dim=1)
        return (inter / (union + 1e-6) * mask).sum() / (mask.sum() + 1e-6)

    def _target_unit_loss(self, logits, labels, mask):
        loss_tmp = self._ce(logits, labels) * mask
        loss = self._masked_mean(loss_tmp, mask, labels.shape[0])
        with torch.no_grad():
            acc = ((logits.argmax(dim=-1) == labels) * mask).sum() / (mask.sum() + 1e-6)
        return {'target_unit_loss': loss, 'target_unit_acc': acc}

    def _target_location_loss(self, logits, labels, mask):
        W = 160
        loss_tmp = self._ce(logits, labels) * mask
        loss = self._masked_mean(loss_tmp, mask, labels.shape[0])
        with torch.no_grad():
            preds = logits.argmax(dim=-1)
            dx = (preds % W) - (labels % W)
            dy = torch.div(preds, W, rounding_mode='floor') - \
                torch.div(labels, W, rounding_mode='floor')
            l2 = ((dx * dx + dy * dy).float().sqrt() * mask).sum() / (mask.sum() + 1e-6)
        return {'target_location_loss': loss, 'target_location_distance_L2': l2}
