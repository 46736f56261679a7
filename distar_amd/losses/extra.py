"""Auxiliary losses for component parity (reference
`ctools/torch_utils/loss/{cross_entropy_loss,multi_logits_loss}.py`):
label-smoothed CE, soft focal loss, and the Hungarian-matching
multi-logits loss."""
import torch
import torch.nn as nn
import torch.nn.functional as F


class LabelSmoothCELoss(nn.Module):
    def __init__(self, ratio=0.1):
        super().__init__()
        self.ratio = ratio

    def forward(self, logits, labels):
        B, N = logits.shape
        val = self.ratio / (N - 1)
        one_hot = torch.full_like(logits, val)
        one_hot.scatter_(1, labels.unsqueeze(1), 1 - self.ratio)
        return -(F.log_softmax(logits, dim=1) * one_hot).sum(dim=1).mean()


class SoftFocalLoss(nn.Module):
    def __init__(self, gamma=2, weight=None, reduction='mean'):
        super().__init__()
        self.gamma = gamma
        self.nll = nn.NLLLoss(weight=weight, reduction=reduction)

    def forward(self, logits, labels):
        p = F.softmax(logits, dim=1)
        return self.nll(((1 - p) ** self.gamma) * torch.log(p + 1e-9), labels)


class MultiLogitsLoss(nn.Module):
    """CE over a bipartite assignment: M logit rows must each pick a distinct
    label; greedy row-wise Hungarian-style matching on the CE cost matrix
    (reference multi_logits_loss.py:30-122, criterion1 path)."""

    def __init__(self, criterion='per_instance', smooth_ratio=0.1):
        super().__init__()
        assert criterion in ('per_instance', 'half_per_instance')
        self.criterion = criterion

    def forward(self, logits, labels):
        """logits (M, N); labels (M,) distinct -> scalar loss."""
        M, N = logits.shape
        assert labels.shape[0] == M
        log_probs = F.log_softmax(logits, dim=1)
        cost = -log_probs[:, labels]              # (M rows, M label slots)
        with torch.no_grad():
            assignment = self._greedy_match(cost)
        picked = cost[torch.arange(M, device=logits.device), assignment]
        return picked.mean()

    @staticmethod
    def _greedy_match(cost):
        M = cost.shape[0]
        cost = cost.clone()
        assignment = torch.zeros(M, dtype=torch.long, device=cost.device)
        used_rows = set()
        used_cols = set()
        for _ in range(M):
            flat = cost.view(-1)
            order = torch.argsort(flat)
            for idx in order.tolist():
                r, c = divmod(idx, M)
                if r not in used_rows and c not in used_cols:
                    assignment[r] = c
                    used_rows.add(r)
                    used_cols.add(c)
                    break
        return assignment
