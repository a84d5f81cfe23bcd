"""Environment registry + ``make_vect_envs``.

Reference parity: ``agilerl/utils/utils.py:222`` (make_vect_envs).  IDs
accept the familiar Gym names and map to the first-party batched envs.
Custom env classes/factories can be registered or passed directly.
"""

from __future__ import annotations

from typing import Callable, Dict, Optional, Union

from .base import VecEnv
from .cartpole import CartPoleVecEnv
from .lunar_lander import LunarLanderContinuousVecEnv, LunarLanderVecEnv
from .classic_control import AcrobotVecEnv, MountainCarContinuousVecEnv, MountainCarVecEnv
from .pendulum import PendulumVecEnv
from .visual import BreakoutLiteVecEnv, CatchPongVecEnv
from .probe import (
    ConstantRewardEnv,
    ConstantRewardContActionsEnv,
    DiscountedRewardEnv,
    FixedObsPolicyContActionsEnv,
    FixedObsPolicyEnv,
    ObsDependentRewardEnv,
    PolicyEnv,
)

__all__ = ["ENV_REGISTRY", "register_env", "make_vect_envs"]

ENV_REGISTRY: Dict[str, Callable[..., VecEnv]] = {
    "CartPole-v1": CartPoleVecEnv,
    "CartPole-v0": CartPoleVecEnv,
    "LunarLander-v2": LunarLanderVecEnv,
    "LunarLander-v3": LunarLanderVecEnv,
    "LunarLanderContinuous-v2": LunarLanderContinuousVecEnv,
    "LunarLanderContinuous-v3": LunarLanderContinuousVecEnv,
    "Pendulum-v1": PendulumVecEnv,
    "MountainCar-v0": MountainCarVecEnv,
    "MountainCarContinuous-v0": MountainCarContinuousVecEnv,
    "Acrobot-v1": AcrobotVecEnv,
    "CatchPong-v0": CatchPongVecEnv,
    "BreakoutLite-v0": BreakoutLiteVecEnv,
    "PongLike-v0": CatchPongVecEnv,
    "probe/ConstantReward": ConstantRewardEnv,
    "probe/ObsDependentReward": ObsDependentRewardEnv,
    "probe/DiscountedReward": DiscountedRewardEnv,
    "probe/FixedObsPolicy": FixedObsPolicyEnv,
    "probe/Policy": PolicyEnv,
    "probe/ConstantRewardContActions": ConstantRewardContActionsEnv,
    "probe/FixedObsPolicyContActions": FixedObsPolicyContActionsEnv,
}


def register_env(env_id: str, factory: Callable[..., VecEnv]) -> None:
    ENV_REGISTRY[env_id] = factory


def make_vect_envs(
    env_id: Optional[Union[str, Callable]] = None,
    num_envs: int = 1,
    seed: Optional[int] = None,
    **env_kwargs,
) -> VecEnv:
    """Create a natively-batched vectorized env."""
    if callable(env_id):
        return env_id(num_envs=num_envs, seed=seed, **env_kwargs)
    if env_id not in ENV_REGISTRY:
        raise KeyError(
            f"Unknown env id '{env_id}'. Registered: {sorted(ENV_REGISTRY)}. "
            "Use register_env() for custom environments."
        )
    return ENV_REGISTRY[env_id](num_envs=num_envs, seed=seed, **env_kwargs)
