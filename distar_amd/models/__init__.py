from .alphastar.model import Model
