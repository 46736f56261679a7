"""Autoregressive policy heads.

Functional parity with the reference's
`distar/agent/default/model/head/{action_type_head,action_arg_head}.py`;
module/key layout matches the reference checkpoints.

MI355X-first restructure (the big one): the reference's SelectedUnitsHead
*training* path runs a Python loop of up to 64 sequential steps, each doing
fc+LSTM+masking on the whole batch (`action_arg_head.py:168-216`).  Under
teacher forcing, both the autoregressive-embedding evolution and the
logits-mask evolution are pure functions of the *labels*, so we compute them
in closed form:

  - cumulative label one-hots / first-occurrence / end-flag gating via
    `cumsum` over the selection axis (no loop),
  - all S lstm inputs with one batched fc pair,
  - the (S, B, 32) query chain with the tiny LN-LSTM (the only truly
    sequential part, 32-wide),
  - the (B, S, N+1) logits with a single batched GEMM against the keys.

This removes ~60 sequential launches x 10 kernels from the SL/RL train step.
The sampling path (data-dependent) keeps the reference's step loop.
"""
import math
from typing import List, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch import Tensor

from ..nn.blocks import (fc_block, conv2d_block, build_activation, ResBlock,
                         ResFCBlock, GatedResBlock, GLU, sequence_mask)
from ..nn.lnlstm import script_lnlstm
from ...lib.consts import MAX_ENTITY_NUM, MAX_SELECTED_UNITS_NUM
from ...lib.stat import ACTION_RACE_MASK


class ActionTypeHead(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.whole_cfg = cfg
        self.cfg = cfg.model.policy.head.action_type_head
        self.act = build_activation(self.cfg.activation)
        self.project = fc_block(self.cfg.input_dim, self.cfg.res_dim, activation=self.act,
                                norm_type=None)
        self.res = nn.Sequential(*[ResFCBlock(self.cfg.res_dim, self.act, self.cfg.norm_type)
                                   for _ in range(self.cfg.res_num)])
        self.action_fc = GLU(self.cfg.res_dim, self.cfg.action_num, self.cfg.context_dim)
        self.action_map_fc1 = fc_block(self.cfg.action_num, self.cfg.action_map_dim,
                                       activation=self.act, norm_type=None)
        self.action_map_fc2 = fc_block(self.cfg.action_map_dim, self.cfg.action_map_dim,
                                       activation=None, norm_type=None)
        self.glu1 = GLU(self.cfg.action_map_dim, self.cfg.gate_dim, self.cfg.context_dim)
        self.glu2 = GLU(self.cfg.input_dim, self.cfg.gate_dim, self.cfg.context_dim)
        self.action_num = self.cfg.action_num
        self.use_mask = cfg.get('common', {}).get('type', 'train') == 'play'
        self.race = 'zerg'

    def forward(self, lstm_output, scalar_context, action_type: Optional[Tensor] = None
                ) -> Tuple[Tensor, Tensor, Tensor]:
        x = self.project(lstm_output)
        x = self.res(x)
        x = self.action_fc(x, scalar_context)
        x = x / self.whole_cfg.model.temperature
        if self.use_mask:
            mask = ACTION_RACE_MASK[self.race].to(x.device)
            x = x.masked_fill(~mask.unsqueeze(0), -1e9)
        if action_type is None:
            p = F.softmax(x, dim=1)
            action_type = torch.multinomial(p, 1)[:, 0]
        action_one_hot = F.one_hot(action_type.long(), self.action_num).float()
        embedding1 = self.action_map_fc1(action_one_hot)
        embedding1 = self.action_map_fc2(embedding1)
        embedding1 = self.glu1(embedding1, scalar_context)
        embedding2 = self.glu2(lstm_output, scalar_context)
        return x, action_type, embedding1 + embedding2


class DelayHead(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.whole_cfg = cfg
        self.cfg = cfg.model.policy.head.delay_head
        self.act = build_activation(self.cfg.activation)
        self.fc1 = fc_block(self.cfg.input_dim, self.cfg.decode_dim, activation=self.act, norm_type=None)
        self.fc2 = fc_block(self.cfg.decode_dim, self.cfg.decode_dim, activation=self.act, norm_type=None)
        self.fc3 = fc_block(self.cfg.decode_dim, self.cfg.delay_dim, activation=None, norm_type=None)
        self.embed_fc1 = fc_block(self.cfg.delay_dim, self.cfg.delay_map_dim, activation=self.act, norm_type=None)
        self.embed_fc2 = fc_block(self.cfg.delay_map_dim, self.cfg.input_dim, activation=None, norm_type=None)
        self.delay_dim = self.cfg.delay_dim

    def forward(self, embedding, delay: Optional[Tensor] = None):
        x = self.fc3(self.fc2(self.fc1(embedding)))
        if delay is None:
            p = F.softmax(x, dim=1)
            delay = torch.multinomial(p, 1)[:, 0]
        delay_one_hot = F.one_hot(delay.long(), self.delay_dim).float()
        embedding_delay = self.embed_fc2(self.embed_fc1(delay_one_hot))
        return x, delay, embedding + embedding_delay


class QueuedHead(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.whole_cfg = cfg
        self.cfg = cfg.model.policy.head.queued_head
        self.act = build_activation(self.cfg.activation)
        self.fc1 = fc_block(self.cfg.input_dim, self.cfg.decode_dim, activation=self.act, norm_type=None)
        self.fc2 = fc_block(self.cfg.decode_dim, self.cfg.decode_dim, activation=self.act, norm_type=None)
        self.fc3 = fc_block(self.cfg.decode_dim, self.cfg.queued_dim, activation=None, norm_type=None)
        self.embed_fc1 = fc_block(self.cfg.queued_dim, self.cfg.queued_map_dim, activation=self.act, norm_type=None)
        self.embed_fc2 = fc_block(self.cfg.queued_map_dim, self.cfg.input_dim, activation=None, norm_type=None)
        self.queued_dim = self.cfg.queued_dim

    def forward(self, embedding, queued: Optional[Tensor] = None):
        x = self.fc3(self.fc2(self.fc1(embedding)))
        x = x / self.whole_cfg.model.temperature
        if queued is None:
            p = F.softmax(x, dim=1)
            queued = torch.multinomial(p, 1)[:, 0]
        queued_one_hot = F.one_hot(queued.long(), self.queued_dim).float()
        embedding_queued = self.embed_fc2(self.embed_fc1(queued_one_hot))
        return x, queued, embedding + embedding_queued


class SelectedUnitsHead(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.whole_cfg = cfg
        self.cfg = cfg.model.policy.head.selected_units_head
        self.act = build_activation(self.cfg.activation)
        self.key_fc = fc_block(self.cfg.entity_embedding_dim, self.cfg.key_dim,
                               activation=None, norm_type=None)
        self.query_fc1 = fc_block(self.cfg.input_dim, self.cfg.func_dim, activation=self.act)
        self.query_fc2 = fc_block(self.cfg.func_dim, self.cfg.key_dim, activation=None)
        self.embed_fc1 = fc_block(self.cfg.key_dim, self.cfg.func_dim, activation=self.act, norm_type=None)
        self.embed_fc2 = fc_block(self.cfg.func_dim, self.cfg.input_dim, activation=None, norm_type=None)
        self.max_select_num = MAX_SELECTED_UNITS_NUM
        self.max_entity_num = MAX_ENTITY_NUM
        self.key_dim = self.cfg.key_dim
        self.num_layers = self.cfg.num_layers
        self.lstm = script_lnlstm(self.cfg.key_dim, self.cfg.hidden_dim, self.cfg.num_layers)
        self.end_embedding = nn.Parameter(torch.empty(1, self.key_dim))
        stdv = 1. / math.sqrt(self.end_embedding.size(1))
        self.end_embedding.data.uniform_(-stdv, stdv)
        self.extra_units = self.whole_cfg.get('agent', {}).get('extra_units', False)
        # free-running rollout during train for the IoU metric (reference
        # action_arg_head.py:173,218-259)
        self.test_iou = self.cfg.get('test_iou', False)
        # 'entity_num'/'constant' are dead paths upstream (the reference's
        # _get_key_mask calls a self.embed_fc that is never constructed,
        # action_arg_head.py:131-136 vs :98-99); the working set is:
        self.reduce_type = self.whole_cfg.model.entity_reduce_type
        assert self.reduce_type in ('selected_units_num', 'attention_pool',
                                    'attention_pool_add_num'), self.reduce_type
        if self.reduce_type == 'attention_pool':
            from ..nn.transformer import AttentionPool
            self.attention_pool = AttentionPool(
                key_dim=self.cfg.key_dim, head_num=2,
                output_dim=self.cfg.input_dim)
        elif self.reduce_type == 'attention_pool_add_num':
            from ..nn.transformer import AttentionPool
            self.attention_pool = AttentionPool(
                key_dim=self.cfg.key_dim, head_num=2,
                output_dim=self.cfg.input_dim,
                max_num=MAX_SELECTED_UNITS_NUM + 1)

    def _get_key_mask(self, entity_embedding, entity_num):
        """Keys (with the learned end-embedding spliced in at index
        entity_num) + availability mask (reference `action_arg_head.py:118-149`)."""
        bs, n = entity_embedding.shape[:2]
        key = self.key_fc(entity_embedding)                       # B, N, k
        key = torch.cat([key, key.new_zeros(bs, 1, self.key_dim)], dim=1)  # B, N+1, k
        flag = torch.ones(bs, n + 1, 1, dtype=torch.bool, device=key.device)
        flag[torch.arange(bs, device=key.device), entity_num] = 0
        key = key * flag + self.end_embedding.squeeze(0) * (~flag)
        key_embeddings = key     # same for selected_units_num + attention_*
        new_entity_num = entity_num + 1      # end slot is a valid position
        mask = sequence_mask(new_entity_num, max_len=n + 1)
        return key, mask, key_embeddings

    def _su_embedding_from_labels(self, key_embeddings, labels, entity_num,
                                  selected_units_num, seq_len):
        """Closed-form replacement of the reference's per-step ae update.

        Returns ae_delta (B, S, input_dim): the embed_fc2(embed_fc1(mean of
        already-selected keys)) term for each step s (s=0 term is defined as
        exactly zero — the reference uses the raw base embedding at step 0).
        """
        bs = labels.shape[0]
        L = labels[:, :seq_len]                                    # B, S
        is_end = L == entity_num.unsqueeze(1)                      # B, S
        # included once the end token has appeared (inclusive) -> excluded
        ended = torch.cummax(is_end.int(), dim=1)[0].bool()        # B, S
        include = ~ended
        # first occurrence of each label (set semantics of the one-hot)
        onehot = torch.zeros(bs, seq_len, key_embeddings.shape[1],
                             dtype=torch.int16, device=L.device)
        onehot.scatter_(2, L.unsqueeze(-1), 1)
        cum = onehot.cumsum(dim=1)                                 # B, S, N+1 (int16)
        seen_before = (cum.gather(2, L.unsqueeze(-1)).squeeze(-1) - 1) > 0
        contrib_gate = (include & ~seen_before).float()            # B, S
        keys_at_labels = key_embeddings.gather(
            1, L.unsqueeze(-1).expand(-1, -1, self.key_dim))       # B, S, k
        contrib = keys_at_labels * contrib_gate.unsqueeze(-1)
        cum_sum = contrib.cumsum(dim=1)                            # B, S, k
        cum_cnt = contrib_gate.cumsum(dim=1)                       # B, S
        # ae at step s uses labels < s: shift right by one
        sum_s = torch.cat([cum_sum.new_zeros(bs, 1, self.key_dim), cum_sum[:, :-1]], dim=1)
        cnt_s = torch.cat([cum_cnt.new_zeros(bs, 1), cum_cnt[:, :-1]], dim=1)
        if self.reduce_type == 'selected_units_num':
            mean_s = sum_s.clone()
            # reference: divide only rows with selected_units_num != 0; guard
            # count==0 (keeps the raw sum, avoiding the reference's 0/0 edge)
            div_rows = (selected_units_num != 0).unsqueeze(1) & (cnt_s > 0)
            mean_s = torch.where(div_rows.unsqueeze(-1), sum_s / cnt_s.clamp(min=1).unsqueeze(-1), sum_s)
            ae_delta = self.embed_fc2(self.embed_fc1(mean_s))      # B, S, input_dim
            ae_delta[:, 0] = 0.                                    # step 0: raw base ae
        else:
            # attention-pool variants (reference action_arg_head.py:201-208):
            # ae_s = base + pool(keys, mask=one-hot of labels < s).  The pool
            # is nonlinear in the mask, so it runs per step over cumulative
            # one-hots (S small); LSTM and logits stay batched.
            gated = onehot.float() * include.float().unsqueeze(-1)
            cum_oh = gated.cumsum(dim=1).clamp(max=1.0)            # B, S, N+1
            deltas = [key_embeddings.new_zeros(bs, self.cfg.input_dim)]
            for st in range(1, seq_len):
                m = cum_oh[:, st - 1].unsqueeze(-1)
                if self.reduce_type == 'attention_pool':
                    deltas.append(self.attention_pool(key_embeddings, mask=m))
                else:
                    deltas.append(self.attention_pool(
                        key_embeddings, num=m.sum(dim=(1, 2)), mask=m))
            ae_delta = torch.stack(deltas, dim=1)                  # B, S, input_dim
        # prev-selected mask for the logits (any label j < s disables index L[b,j])
        prev_cnt = torch.cat([cum.new_zeros(bs, 1, cum.shape[2]), cum[:, :-1]], dim=1)
        prev_selected = prev_cnt > 0                               # B, S, N+1
        # final ae after the full unroll (consumed by the target-unit head)
        if self.reduce_type == 'selected_units_num':
            fin_div = (selected_units_num != 0) & (cum_cnt[:, -1] > 0)
            fin_mean = torch.where(fin_div.unsqueeze(-1),
                                   cum_sum[:, -1] / cum_cnt[:, -1].clamp(min=1).unsqueeze(-1),
                                   cum_sum[:, -1])
            final_delta = self.embed_fc2(self.embed_fc1(fin_mean))
        else:
            mfin = cum_oh[:, -1].unsqueeze(-1)
            if self.reduce_type == 'attention_pool':
                final_delta = self.attention_pool(key_embeddings, mask=mfin)
            else:
                final_delta = self.attention_pool(
                    key_embeddings, num=mfin.sum(dim=(1, 2)), mask=mfin)
        return ae_delta, prev_selected, final_delta

    def _query_train(self, key, entity_num, autoregressive_embedding, logits_mask,
                     key_embeddings, selected_units_num, selected_units):
        bs = autoregressive_embedding.shape[0]
        seq_len = max(int(selected_units_num.max()), 1)
        ae_delta, prev_selected, final_delta = self._su_embedding_from_labels(
            key_embeddings, selected_units, entity_num, selected_units_num, seq_len)
        ae_all = autoregressive_embedding.unsqueeze(1) + ae_delta          # B, S, D
        lstm_input = self.query_fc2(self.query_fc1(ae_all))                # B, S, k
        lstm_input = lstm_input.transpose(0, 1).contiguous()               # S, B, k
        state = [(lstm_input.new_zeros(bs, self.cfg.hidden_dim),
                  lstm_input.new_zeros(bs, self.cfg.hidden_dim))
                 for _ in range(self.num_layers)]
        queries, _ = self.lstm(lstm_input, state)                          # S, B, k
        logits = torch.einsum('sbk,bnk->bsn', queries, key)                # B, S, N+1
        # mask: availability, minus previously-selected, end slot from step 1
        arange = torch.arange(bs, device=key.device)
        mask = logits_mask.unsqueeze(1) & ~prev_selected                   # B, S, N+1
        mask[arange, 0, entity_num] = False
        logits = logits.masked_fill(~mask, -1e9)
        final_ae = autoregressive_embedding + final_delta
        results = None
        if self.test_iou:
            results = self._iou_rollout(key, entity_num, autoregressive_embedding,
                                        logits_mask, key_embeddings, seq_len)
        return logits, results, final_ae, selected_units_num, results

    def _iou_rollout(self, key, entity_num, autoregressive_embedding,
                     logits_mask, key_embeddings, seq_len):
        """Free-running greedy-sampled selections capped at the teacher
        sequence length (reference action_arg_head.py:218-259); used only for
        the SL IoU metric, no gradients."""
        with torch.no_grad():
            bs = autoregressive_embedding.shape[0]
            device = autoregressive_embedding.device
            ae = autoregressive_embedding
            end_flag = torch.zeros(bs, dtype=torch.bool, device=device)
            mask = logits_mask.clone()
            arange = torch.arange(bs, device=device)
            mask[arange, entity_num] = False
            state = [(ae.new_zeros(bs, self.cfg.hidden_dim),
                      ae.new_zeros(bs, self.cfg.hidden_dim))
                     for _ in range(self.num_layers)]
            sel_oh = ae.new_zeros(bs, key.shape[1], 1)
            results = []
            result = None
            for i in range(seq_len):
                if i == 1:
                    mask[arange, entity_num] = True
                if result is not None:
                    mask[arange, result] = False
                lstm_input = self.query_fc2(self.query_fc1(ae)).unsqueeze(0)
                lstm_output, state = self.lstm(lstm_input, state)
                step_logits = (lstm_output.permute(1, 0, 2) * key).sum(dim=2)
                step_logits = step_logits.masked_fill(~mask, -1e9)
                result = torch.multinomial(F.softmax(step_logits, dim=-1), 1)[:, 0]
                end_flag[result == entity_num] = True
                results.append(result)
                sel_oh = sel_oh.clone()
                sel_oh[arange[~end_flag], result[~end_flag]] = 1
                if self.reduce_type == 'selected_units_num':
                    emb = (key_embeddings * sel_oh).sum(dim=1)
                    cnt = sel_oh.sum(dim=1)
                    nz = (cnt > 0).squeeze(-1)
                    emb[nz] = emb[nz] / cnt[nz]
                    ae = autoregressive_embedding + self.embed_fc2(self.embed_fc1(emb))
                elif self.reduce_type == 'attention_pool':
                    ae = autoregressive_embedding + \
                        self.attention_pool(key_embeddings, mask=sel_oh)
                else:
                    ae = autoregressive_embedding + self.attention_pool(
                        key_embeddings, num=sel_oh.sum(dim=1).squeeze(-1),
                        mask=sel_oh)
            return torch.stack(results, dim=0).transpose(1, 0).contiguous()

    def _query_sample_hip(self, key, entity_num, autoregressive_embedding,
                          logits_mask, key_embeddings, su_mask, uniforms=None):
        """K7: the whole data-dependent loop as ONE HIP kernel
        (ops/hip/su_sample.hip); final ae recomputed host-side from the
        selected keys (cheap batched fc pair)."""
        from ...ops.su_sample import su_sample
        bs = autoregressive_embedding.shape[0]
        device = autoregressive_embedding.device
        logits, results, num = su_sample(
            self, autoregressive_embedding.float(), key, logits_mask, su_mask,
            entity_num, self.whole_cfg.model.temperature, uniforms=uniforms)
        S = max(int(num.max()), 1)
        logits = logits[:, :S]
        results = results[:, :S]
        # final ae = base + embed(mean of selected keys), zero-selection rows
        # keep the raw (zero) sum like the reference
        arange = torch.arange(bs, device=device)
        sel_mask = (torch.arange(S, device=device).unsqueeze(0) <
                    (num - 1).clamp(min=0).unsqueeze(1))
        gathered = key.gather(1, results.clamp(min=0).unsqueeze(-1)
                              .expand(-1, -1, self.key_dim))
        sel_sum = (gathered * sel_mask.unsqueeze(-1)).sum(dim=1)
        cnt = sel_mask.sum(dim=1, keepdim=True).float()
        mean = torch.where(cnt > 0, sel_sum / cnt.clamp(min=1), sel_sum)
        ae = autoregressive_embedding + self.embed_fc2(self.embed_fc1(mean))
        extra_units = torch.zeros(bs, MAX_ENTITY_NUM + 1, device=device)
        if self.extra_units:
            # reference action_arg_head.py:307-309: extras come from the LAST
            # executed step's logits, and only for rows whose selection was
            # truncated at the cap (rows that picked the end token get none)
            last = (num - 1).clamp(min=0)
            last_logits = logits[arange, last]
            end_logit = last_logits[arange, entity_num]
            ended = (results.gather(1, last.unsqueeze(1)).squeeze(1) ==
                     entity_num) | (num == 0)
            extra_units[:, :last_logits.shape[1]] = \
                ((last_logits > end_logit.unsqueeze(1)) &
                 ~ended.unsqueeze(1)).float()
        return logits, results, ae, num, extra_units

    def _query_sample(self, key, entity_num, autoregressive_embedding, logits_mask,
                      key_embeddings, su_mask, uniforms=None):
        """Data-dependent sampling loop (reference `action_arg_head.py:262-313`)."""
        import os
        if autoregressive_embedding.is_cuda and \
                self.reduce_type == 'selected_units_num' and \
                os.environ.get('DISTAR_AMD_DISABLE_HIP') != '1':
            return self._query_sample_hip(key, entity_num, autoregressive_embedding,
                                          logits_mask, key_embeddings, su_mask,
                                          uniforms)
        ae = autoregressive_embedding
        bs = ae.shape[0]
        device = ae.device
        end_flag = torch.zeros(bs, dtype=torch.bool, device=device)
        results_list, logits_list = [], []
        state = [(ae.new_zeros(bs, self.cfg.hidden_dim), ae.new_zeros(bs, self.cfg.hidden_dim))
                 for _ in range(self.num_layers)]
        arange = torch.arange(bs, device=device)
        logits_mask = logits_mask.clone()
        logits_mask[arange, entity_num] = False
        selected_units_num = torch.full((bs,), self.max_select_num, dtype=torch.long, device=device)
        end_flag[~su_mask] = True
        selected_units_num[~su_mask] = 0
        result: Optional[Tensor] = None
        sel_sum = ae.new_zeros(bs, self.key_dim)
        sel_cnt = ae.new_zeros(bs)
        sel_oh = ae.new_zeros(bs, key.shape[1])
        step_logits = None
        for i in range(self.max_select_num):
            if i == 1:
                logits_mask[arange, entity_num] = True     # end flag selectable now
            if result is not None:
                logits_mask[arange, result.detach()] = False
            lstm_input = self.query_fc2(self.query_fc1(ae)).unsqueeze(0)
            lstm_output, state = self.lstm(lstm_input, state)
            queries = lstm_output.permute(1, 0, 2)                      # B, 1, k
            step_logits = (queries * key).sum(dim=2)                    # B, N+1
            step_logits = step_logits.masked_fill(~logits_mask, -1e9)
            p = F.softmax(step_logits / self.whole_cfg.model.temperature, dim=-1)
            if uniforms is None:
                units = torch.multinomial(p, 1)[:, 0]
            else:   # deterministic inverse-CDF (golden tests vs the kernel)
                cdf = p.cumsum(-1)
                target = uniforms[:, i:i + 1] * cdf[:, -1:]
                units = torch.searchsorted(cdf, target).clamp(
                    max=cdf.shape[1] - 1)[:, 0]
            result = units
            newly_ended = (result == entity_num) & ~end_flag
            selected_units_num[newly_ended] = i + 1
            end_flag[result == entity_num] = True
            results_list.append(result)
            logits_list.append(step_logits)
            picked = (~end_flag).float()
            if self.reduce_type == 'selected_units_num':
                sel_sum = sel_sum + key_embeddings[arange, result] * picked.unsqueeze(1)
                sel_cnt = sel_cnt + picked
                mean = torch.where((sel_cnt > 0).unsqueeze(1),
                                   sel_sum / sel_cnt.clamp(min=1).unsqueeze(1),
                                   sel_sum)
                ae = autoregressive_embedding + self.embed_fc2(self.embed_fc1(mean))
            else:
                sel_oh = sel_oh.clone()
                sel_oh[arange[~end_flag], result[~end_flag]] = 1.
                m = sel_oh.unsqueeze(-1)
                if self.reduce_type == 'attention_pool':
                    ae = autoregressive_embedding + \
                        self.attention_pool(key_embeddings, mask=m)
                else:
                    ae = autoregressive_embedding + self.attention_pool(
                        key_embeddings, num=sel_oh.sum(dim=1), mask=m)
            if bool(end_flag.all()):
                break
        extra_units = torch.zeros(bs, MAX_ENTITY_NUM + 1, device=device)
        if self.extra_units and step_logits is not None:
            end_flag_logit = step_logits[arange, entity_num]
            extra_units = ((step_logits > end_flag_logit.unsqueeze(1)) & ~end_flag.unsqueeze(1)).float()
        results = torch.stack(results_list, dim=0).transpose(1, 0).contiguous()
        logits = torch.stack(logits_list, dim=0).transpose(1, 0).contiguous()
        return logits, results, ae, selected_units_num, extra_units

    def forward(self, embedding, entity_embedding, entity_num,
                selected_units_num: Optional[Tensor] = None,
                selected_units: Optional[Tensor] = None,
                su_mask: Optional[Tensor] = None):
        key, mask, key_embeddings = self._get_key_mask(entity_embedding, entity_num)
        if selected_units is not None and selected_units_num is not None:
            return self._query_train(key, entity_num, embedding, mask, key_embeddings,
                                     selected_units_num, selected_units)
        return self._query_sample(key, entity_num, embedding, mask, key_embeddings, su_mask)


class TargetUnitHead(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.whole_cfg = cfg
        self.cfg = cfg.model.policy.head.target_unit_head
        self.act = build_activation(self.cfg.activation)
        self.key_fc = fc_block(self.cfg.entity_embedding_dim, self.cfg.key_dim,
                               activation=None, norm_type=None)
        self.query_fc1 = fc_block(self.cfg.input_dim, self.cfg.key_dim, activation=self.act,
                                  norm_type=None)
        self.query_fc2 = fc_block(self.cfg.key_dim, self.cfg.key_dim, activation=None,
                                  norm_type=None)
        self.key_dim = self.cfg.key_dim
        self.max_entity_num = MAX_ENTITY_NUM

    def forward(self, embedding, entity_embedding, entity_num,
                target_unit: Optional[Tensor] = None):
        key = self.key_fc(entity_embedding)
        mask = sequence_mask(entity_num, max_len=entity_embedding.shape[1])
        query = self.query_fc2(self.query_fc1(embedding))
        logits = (query.unsqueeze(1) * key).sum(dim=2)
        logits = logits.masked_fill(~mask, -1e9)
        logits = logits / self.whole_cfg.model.temperature
        if target_unit is None:
            p = F.softmax(logits, dim=1)
            target_unit = torch.multinomial(p, 1)[:, 0]
        return logits, target_unit


class LocationHead(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.whole_cfg = cfg
        self.cfg = cfg.model.policy.head.location_head
        self.act = build_activation(self.cfg.activation)
        self.reshape_size = self.cfg.reshape_size
        self.reshape_channel = self.cfg.reshape_channel
        self.conv1 = conv2d_block(self.cfg.map_skip_dim + self.cfg.reshape_channel,
                                  self.cfg.res_dim, 1, 1, 0,
                                  activation=build_activation(self.cfg.activation),
                                  norm_type=None)
        self.res_dim = self.cfg.res_dim
        self.use_film = self.cfg.get('film', False)
        self.use_gate = self.cfg.get('gate', False)
        self.use_unet = self.cfg.get('unet', False)
        spatial_y = self.whole_cfg.model.spatial_y
        spatial_x = self.whole_cfg.model.spatial_x
        self.project_embed = fc_block(
            self.cfg.input_dim, (spatial_y // 8) * (spatial_x // 8) * 4,
            activation=build_activation(self.cfg.activation))
        if self.use_film:
            # FiLM conditioning of each res stage on the LSTM embedding
            # (reference action_arg_head.py:385-402)
            from ..nn.blocks import FiLMedResBlock
            self.film_fc = fc_block(self.cfg.input_dim, self.res_dim,
                                    activation=build_activation(self.cfg.activation))
            self.film_gamma = nn.ModuleList()
            self.film_beta = nn.ModuleList()
            self.film = nn.ModuleList()
        self.res = nn.ModuleList()
        for _ in range(self.cfg.res_num):
            if self.use_gate:
                self.res.append(GatedResBlock(self.res_dim, self.res_dim, 3, 1, 1,
                                              activation=build_activation(self.cfg.activation),
                                              norm_type=None))
            else:
                self.res.append(ResBlock(self.res_dim, build_activation(self.cfg.activation),
                                         norm_type=None))
            if self.use_film:
                g = nn.Linear(self.res_dim, self.res_dim)
                b = nn.Linear(self.res_dim, self.res_dim)
                nn.init.xavier_uniform_(g.weight)
                nn.init.xavier_uniform_(b.weight)
                self.film_gamma.append(g)
                self.film_beta.append(b)
                self.film.append(FiLMedResBlock(self.res_dim, with_cond=[True]))
        self.upsample = nn.ModuleList()
        dims = [self.res_dim] + list(self.cfg.upsample_dims)
        assert self.cfg.upsample_type in ('deconv', 'nearest', 'bilinear')
        from ..nn.blocks import deconv2d_block
        for i in range(len(self.cfg.upsample_dims)):
            activation = None if i == len(self.cfg.upsample_dims) - 1 \
                else build_activation(self.cfg.activation)
            if self.cfg.upsample_type == 'deconv':
                self.upsample.append(deconv2d_block(dims[i], dims[i + 1], 4, 2, 1,
                                                    activation=activation, norm_type=None))
            else:
                self.upsample.append(conv2d_block(dims[i], dims[i + 1], 3, 1, 1,
                                                  activation=activation, norm_type=None))

    def forward(self, embedding, map_skip: List[Tensor], location: Optional[Tensor] = None):
        spatial_y = self.whole_cfg.model.spatial_y
        spatial_x = self.whole_cfg.model.spatial_x
        projected = self.project_embed(embedding)
        reshaped = projected.reshape(projected.shape[0], self.reshape_channel,
                                     spatial_y // 8, spatial_x // 8)
        cat_feature = torch.cat([reshaped, map_skip[-1]], dim=1)
        x = self.conv1(self.act(cat_feature))
        film_embedding = self.film_fc(embedding) if self.use_film else None
        for i in range(len(self.res)):
            x = x + map_skip[len(map_skip) - i - 1]
            x = self.res[i](x, x) if self.use_gate else self.res[i](x)
            if self.use_film:
                x = self.film[i](x, gammas=self.film_gamma[i](film_embedding),
                                 betas=self.film_beta[i](film_embedding))
        from ...ops.upsample import upsample2x_bilinear
        for i, layer in enumerate(self.upsample):
            if self.cfg.upsample_type == 'nearest':
                x = F.interpolate(x, scale_factor=2., mode='nearest')
            elif self.cfg.upsample_type == 'bilinear':
                x = upsample2x_bilinear(x)
            if self.use_unet:
                x = x + map_skip[len(map_skip) - len(self.res) - i - 1]
            x = layer(x)
        logits_flat = x.reshape(x.shape[0], -1) / self.whole_cfg.model.temperature
        if location is None:
            p = F.softmax(logits_flat, dim=1)
            location = torch.multinomial(p, 1)[:, 0]
        return logits_flat, location
