"""(T, B) reverse scans: V-trace values and generalized lambda returns.

The scans are sequential in T but embarrassingly parallel across B (and
heads/baselines); the MI355X path runs each scan as ONE HIP kernel with one
thread per batch column instead of the reference's
T-iteration Python loop (`as_rl_utils.py:157-218,284-312`).  The eager
fallback below is the numerics oracle the kernel is golden-tested against.

All scans run under no_grad by their callers (advantages/returns are
treated as constants in the losses), so no backward is needed.
"""
import torch

from . import hip_ext


def _lambda_return_scan_eager(rewards, gammas, bootstrap_values_tp1, lambda_):
    result = torch.empty_like(rewards)
    T = rewards.shape[0]
    result[T - 1] = rewards[T - 1] + gammas[T - 1] * bootstrap_values_tp1[T - 1]
    discounts = gammas * lambda_
    for t in reversed(range(T - 1)):
        result[t] = rewards[t] + discounts[t] * result[t + 1] \
            + (gammas[t] - discounts[t]) * bootstrap_values_tp1[t]
    return result


def _vtrace_scan_eager(clipped_rhos, clipped_cs, rewards, bootstrap_values,
                       gammas, lambda_):
    deltas = clipped_rhos * (rewards + gammas * bootstrap_values[1:]
                             - bootstrap_values[:-1])
    vtrace_val = torch.empty_like(bootstrap_values)
    vtrace_val[-1] = bootstrap_values[-1]
    T = rewards.shape[0]
    for t in reversed(range(T)):
        vtrace_val[t] = bootstrap_values[t] + deltas[t] \
            + gammas[t] * lambda_[t] * clipped_cs[t] \
            * (vtrace_val[t + 1] - bootstrap_values[t + 1])
    return vtrace_val


def lambda_return_scan(rewards, gammas, bootstrap_values_tp1, lambda_):
    """result: (T, B); bootstrap_values_tp1 holds V_{t+1} for t in [0, T)."""
    ext = hip_ext.maybe_ext(rewards)
    if ext is not None:
        return ext.lambda_return_scan(
            rewards.contiguous().float(), gammas.contiguous().float(),
            bootstrap_values_tp1.contiguous().float(), lambda_.contiguous().float())
    return _lambda_return_scan_eager(rewards, gammas, bootstrap_values_tp1, lambda_)


def vtrace_scan(clipped_rhos, clipped_cs, rewards, bootstrap_values, gammas, lambda_):
    """vtrace_val: (T+1, B)."""
    ext = hip_ext.maybe_ext(rewards)
    if ext is not None:
        return ext.vtrace_scan(
            clipped_rhos.contiguous().float(), clipped_cs.contiguous().float(),
            rewards.contiguous().float(), bootstrap_values.contiguous().float(),
            gammas.contiguous().float(), lambda_.contiguous().float())
    return _vtrace_scan_eager(clipped_rhos, clipped_cs, rewards, bootstrap_values,
                              gammas, lambda_)
