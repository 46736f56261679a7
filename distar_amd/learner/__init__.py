from .base_learner import BaseLearner
from .sl_learner import SLLearner
from .rl_learner import RLLearner
