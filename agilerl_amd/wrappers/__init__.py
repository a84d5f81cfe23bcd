from .agent import AgentWrapper, RSNorm, AsyncAgentsWrapper
from .learning import Skill, BanditEnv
from .make_evolvable import MakeEvolvable
from .pettingzoo_wrappers import AutoResetParallelWrapper

__all__ = ["AgentWrapper", "RSNorm", "AsyncAgentsWrapper", "Skill", "BanditEnv", "MakeEvolvable", "AutoResetParallelWrapper"]
