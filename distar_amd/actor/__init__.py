from .agent import Agent
from .actor import Actor
from .comm import ActorComm, LearnerComm
