"""Prioritized fictitious self-play opponent sampling (reference
`ctools/worker/league/algorithms.py:58-86`)."""
import numpy as np


def pfsp(win_rates: np.ndarray, weighting: str = 'variance') -> np.ndarray:
    """Win rates vs N opponents -> selection probabilities."""
    weighting_func = {
        'squared': lambda x: (1 - x) ** 2,
        'variance': lambda x: x * (1 - x),
        'normal': lambda x: np.minimum(0.5, 1 - x),
    }
    if weighting not in weighting_func:
        raise KeyError(f'invalid pfsp weighting: {weighting}')
    assert isinstance(win_rates, np.ndarray) and win_rates.shape[0] >= 1
    if win_rates.sum() < 1e-8:
        return np.full_like(win_rates, 1.0 / len(win_rates), dtype=np.float64)
    w = weighting_func[weighting](win_rates.astype(np.float64))
    if w.sum() < 1e-8:
        return np.full_like(win_rates, 1.0 / len(win_rates), dtype=np.float64)
    return w / w.sum()
