"""Autoregressive head chain (reference `model/policy.py`)."""
from typing import Dict, List

import torch
import torch.nn as nn
from torch import Tensor

from .heads import (ActionTypeHead, DelayHead, QueuedHead, SelectedUnitsHead,
                    TargetUnitHead, LocationHead)
from ...lib.actions import SELECTED_UNITS_MASK


class Policy(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.whole_cfg = cfg
        self.cfg = cfg.model.policy
        self.action_type_head = ActionTypeHead(cfg)
        self.delay_head = DelayHead(cfg)
        self.queued_head = QueuedHead(cfg)
        self.selected_units_head = SelectedUnitsHead(cfg)
        self.target_unit_head = TargetUnitHead(cfg)
        self.location_head = LocationHead(cfg)

    def forward(self, lstm_output: Tensor, entity_embeddings: Tensor,
                map_skip: List[Tensor], scalar_context: Tensor, entity_num: Tensor):
        action: Dict[str, Tensor] = {}
        logit: Dict[str, Tensor] = {}
        logit['action_type'], action['action_type'], embeddings = \
            self.action_type_head(lstm_output, scalar_context)
        logit['delay'], action['delay'], embeddings = self.delay_head(embeddings)
        logit['queued'], action['queued'], embeddings = self.queued_head(embeddings)
        su_mask = SELECTED_UNITS_MASK.to(action['action_type'].device)[action['action_type']]
        logit['selected_units'], action['selected_units'], embeddings, selected_units_num, extra_units = \
            self.selected_units_head(embeddings, entity_embeddings, entity_num, None, None, su_mask)
        logit['target_unit'], action['target_unit'] = \
            self.target_unit_head(embeddings, entity_embeddings, entity_num)
        logit['target_location'], action['target_location'] = \
            self.location_head(embeddings, map_skip)
        return action, selected_units_num, logit, extra_units

    def train_forward(self, lstm_output, entity_embeddings, map_skip: List[Tensor],
                      scalar_context, entity_num, action_info, selected_units_num):
        action: Dict[str, Tensor] = {}
        logit: Dict[str, Tensor] = {}
        logit['action_type'], action['action_type'], embeddings = \
            self.action_type_head(lstm_output, scalar_context, action_info['action_type'])
        logit['delay'], action['delay'], embeddings = \
            self.delay_head(embeddings, action_info['delay'])
        logit['queued'], action['queued'], embeddings = \
            self.queued_head(embeddings, action_info['queued'])
        logit['selected_units'], action['selected_units'], embeddings, selected_units_num, _ = \
            self.selected_units_head(embeddings, entity_embeddings, entity_num,
                                     selected_units_num, action_info['selected_units'])
        logit['target_unit'], action['target_unit'] = \
            self.target_unit_head(embeddings, entity_embeddings, entity_num,
                                  action_info['target_unit'])
        logit['target_location'], action['target_location'] = \
            self.location_head(embeddings, map_skip, action_info['target_location'])
        return action, selected_units_num, logit
