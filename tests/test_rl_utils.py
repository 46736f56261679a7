"""Numerics tests for the RL scan/return math (reference as_rl_utils.py)."""
import torch

from distar_amd.losses import rl_utils
from distar_amd.ops import scans


def manual_lambda_return(rewards, gammas, v_tp1, lam):
    T = rewards.shape[0]
    out = torch.empty_like(rewards)
    out[-1] = rewards[-1] + gammas[-1] * v_tp1[-1]
    for t in reversed(range(T - 1)):
        out[t] = rewards[t] + gammas[t] * (lam[t] * out[t + 1] + (1 - lam[t]) * v_tp1[t])
    return out


def test_lambda_return_scan_matches_manual():
    torch.manual_seed(0)
    T, B = 16, 5
    r, g, v, lam = torch.randn(T, B), torch.rand(T, B), torch.randn(T, B), torch.rand(T, B)
    torch.testing.assert_close(scans._lambda_return_scan_eager(r, g, v, lam),
                               manual_lambda_return(r, g, v, lam), rtol=1e-5, atol=1e-5)


def test_vtrace_values_properties():
    torch.manual_seed(1)
    T, B = 8, 4
    rhos = torch.rand(T, B)
    cs = torch.rand(T, B)
    r = torch.randn(T, B)
    v = torch.randn(T + 1, B)
    g = torch.ones(T, B)
    lam = torch.ones(T, B)
    vt = scans._vtrace_scan_eager(rhos, cs, r, v, g, lam)
    assert vt.shape == (T + 1, B)
    torch.testing.assert_close(vt[-1], v[-1])
    # with rho=c=1 and gamma=lambda=1, v-trace equals the Monte-Carlo return
    ones = torch.ones(T, B)
    vt_full = scans._vtrace_scan_eager(ones, ones, r, v, g, lam)
    mc = torch.flip(torch.cumsum(torch.flip(r, [0]), 0), [0]) + v[-1]
    torch.testing.assert_close(vt_full[:-1], mc, rtol=1e-4, atol=1e-4)


def test_upgo_returns_shape_and_terminal():
    torch.manual_seed(2)
    T, B = 6, 3
    r = torch.randn(T, B)
    v = torch.randn(T + 1, B)
    ret = rl_utils.upgo_returns(r, v)
    assert ret.shape == (T, B)
    torch.testing.assert_close(ret[-1], r[-1] + v[-1])


def test_td_lambda_loss_zero_when_perfect():
    T, B = 5, 2
    r = torch.zeros(T, B)
    v = torch.zeros(T + 1, B)
    loss = rl_utils.td_lambda_loss(v, r)
    assert float(loss) == 0.0


def test_entropy_of_uniform_is_one():
    """Normalized entropy of a uniform categorical is 1."""
    T, B, N = 3, 2, 7
    probs = torch.full((T, B, N), 1.0 / N)
    log_probs = probs.log()
    mask = {'actions_mask': {k: torch.ones(T, B) for k in rl_utils.HEAD_TYPES}}
    d = {k: probs for k in ['action_type', 'delay']}
    ld = {k: log_probs for k in ['action_type', 'delay']}
    ent = -(probs * log_probs).sum(-1) / torch.log(torch.tensor([float(N)]))
    torch.testing.assert_close(ent.mean(), torch.tensor(1.0), rtol=1e-5, atol=1e-5)
