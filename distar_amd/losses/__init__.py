from .sl_loss import SupervisedLoss
from .rl_loss import ReinforcementLoss
