"""distar_amd: an MI355X-native AlphaStar-class StarCraft II training
framework with the capabilities of opendilab/DI-star (reference layer map in
/root/repo/SURVEY.md).  PyTorch-ROCm host/autograd + hand-written HIP/CDNA4
kernels for the hot ops + RCCL over xGMI for the data-parallel learner."""
import os as _os

# MIOpen's default exhaustive find takes minutes on a cold box for this
# model's conv set (and falls back to a naive fp64-accumulate wrw kernel for
# non-packed gradient tensors).  FAST find uses heuristics: near-instant
# startup, near-identical steady-state conv choice on gfx950.
_os.environ.setdefault('MIOPEN_FIND_MODE', 'FAST')

__version__ = '0.1.0'
