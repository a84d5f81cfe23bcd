from .base import VecEnv, BatchedVecEnv
from .cartpole import CartPoleVecEnv
from .lunar_lander import LunarLanderVecEnv
from .pendulum import PendulumVecEnv
from .visual import CatchPongVecEnv
from .registry import ENV_REGISTRY, make_vect_envs, register_env
from . import probe

__all__ = [
    "VecEnv",
    "BatchedVecEnv",
    "CartPoleVecEnv",
    "LunarLanderVecEnv",
    "PendulumVecEnv",
    "CatchPongVecEnv",
    "ENV_REGISTRY",
    "make_vect_envs",
    "register_env",
    "probe",
]
