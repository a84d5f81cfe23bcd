from .dqn import DQN
from .dqn_rainbow import RainbowDQN
from .cqn import CQN
from .ddpg import DDPG
from .td3 import TD3
from .ppo import PPO
from .maddpg import MADDPG
from .matd3 import MATD3
from .ippo import IPPO
from .neural_ucb import NeuralUCB
from .neural_ts import NeuralTS

__all__ = [
    "ILQL",
    "BC_LM",
    "DQN",
    "RainbowDQN",
    "CQN",
    "DDPG",
    "TD3",
    "PPO",
    "MADDPG",
    "MATD3",
    "IPPO",
    "NeuralUCB",
    "NeuralTS",
]

from .ilql import ILQL  # noqa: E402,F401
from .bc_lm import BC_LM  # noqa: E402,F401
