from .model import Model, alphastar_model_default_config
