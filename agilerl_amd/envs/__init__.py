from .base import VecEnv, BatchedVecEnv
from .cartpole import CartPoleVecEnv
from .lunar_lander import LunarLanderVecEnv
from .pendulum import PendulumVecEnv
from .classic_control import AcrobotVecEnv, MountainCarContinuousVecEnv, MountainCarVecEnv
from .visual import CatchPongVecEnv
from .registry import ENV_REGISTRY, make_vect_envs, register_env
from . import probe

__all__ = [
    "VecEnv",
    "BatchedVecEnv",
    "CartPoleVecEnv",
    "LunarLanderVecEnv",
    "PendulumVecEnv",
    "MountainCarVecEnv",
    "MountainCarContinuousVecEnv",
    "AcrobotVecEnv",
    "CatchPongVecEnv",
    "ENV_REGISTRY",
    "make_vect_envs",
    "register_env",
    "probe",
]
