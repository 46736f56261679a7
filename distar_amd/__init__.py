"""distar_amd: an MI355X-native AlphaStar-class StarCraft II training
framework with the capabilities of opendilab/DI-star (reference layer map in
/root/repo/SURVEY.md).  PyTorch-ROCm host/autograd + hand-written HIP/CDNA4
kernels for the hot ops + RCCL over xGMI for the data-parallel learner."""
__version__ = '0.1.0'
