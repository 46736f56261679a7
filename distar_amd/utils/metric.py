"""Distance metrics for the pseudo-rewards (reference `ctools/utils/metric.py`):
Levenshtein DP with optional per-position extra cost (build-order reward),
Hamming distance (cumulative-stat reward), flat-location L2."""
import torch


def levenshtein_distance(pred, target, pred_extra=None, target_extra=None,
                         extra_fn=None):
    """O(N*M) DP; when ``extra_fn`` is given, equal elements contribute
    extra_fn(pred_extra[i], target_extra[j]) instead of 0 (the reference's
    location-aware build-order distance)."""
    assert isinstance(pred, torch.Tensor) and isinstance(target, torch.Tensor)
    assert pred.dtype == target.dtype
    n, m = pred.shape[0], target.shape[0]
    if n == 0 or m == 0:
        return torch.as_tensor(max(n, m), dtype=torch.float)
    dp = torch.zeros(n + 1, m + 1)
    dp[0, :] = torch.arange(0, m + 1)
    dp[:, 0] = torch.arange(0, n + 1)
    for i in range(1, n + 1):
        for j in range(1, m + 1):
            if pred[i - 1] == target[j - 1]:
                if extra_fn is not None:
                    sub = dp[i - 1, j - 1] + extra_fn(pred_extra[i - 1],
                                                      target_extra[j - 1])
                else:
                    sub = dp[i - 1, j - 1]
            else:
                sub = dp[i - 1, j - 1] + 1
            dp[i, j] = torch.min(torch.stack([sub, dp[i - 1, j] + 1, dp[i, j - 1] + 1]))
    return dp[n, m].float()


def hamming_distance(pred, target, weight=1.):
    """Per-row Hamming distance of binary tensors (B, N) -> (B,)."""
    assert pred.shape == target.shape
    return ((pred != target) * weight).sum(dim=tuple(range(1, pred.dim()))).float() \
        if pred.dim() > 1 else ((pred != target) * weight).sum().float()


def l2_distance(a, b, min_val=0., max_val=0.8, spatial_x=160):
    """Normalized L2 between flat map locations (reference metric.l2_distance)."""
    x0, y0 = a % spatial_x, a // spatial_x
    x1, y1 = b % spatial_x, b // spatial_x
    d = ((x0 - x1).float() ** 2 + (y0 - y1).float() ** 2).sqrt()
    cost = (d / 10).clamp(min=min_val, max=max_val)
    return cost
