from .agent import AgentWrapper, RSNorm, AsyncAgentsWrapper
from .learning import Skill, BanditEnv
from .make_evolvable import MakeEvolvable

__all__ = ["AgentWrapper", "RSNorm", "AsyncAgentsWrapper", "Skill", "BanditEnv", "MakeEvolvable"]
