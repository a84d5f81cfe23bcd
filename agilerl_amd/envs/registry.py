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
from .toy_text import BlackjackVecEnv
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
    "Blackjack-v1": BlackjackVecEnv,
    # ALE ROMs are unavailable offline; the first-party BreakoutLite env
    # stands in with the same visual-obs pipeline shape
    "ALE/Breakout-v5": BreakoutLiteVecEnv,
    "ALE/Pong-v5": CatchPongVecEnv,
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
    make_env: Optional[Callable] = None,
    should_async_vector: bool = False,
    extra_wrappers=None,
    env_name: Optional[str] = None,
    **env_kwargs,
) -> VecEnv:
    """Create a vectorized env.

    Default path: the first-party *natively batched* registry envs (one
    tensor op steps all N copies — no subprocess fan-out needed).
    Reference-compat paths (utils.py:222): ``make_env`` vectorizes a
    user per-env factory via subprocess workers when
    ``should_async_vector`` (shared-memory AsyncVectorEnv) or a simple
    serial loop otherwise; ``extra_wrappers`` wrap each instance.
    """
    if env_name is not None and env_id is None:
        env_id = env_name  # reference spelling
    if make_env is not None:
        factory = make_env
        if extra_wrappers:
            inner = factory

            def factory():
                e = inner()
                for wrapper_cls in extra_wrappers:
                    e = wrapper_cls(e)
                return e

        from ..vector.async_vec_env import AsyncVectorEnv

        if should_async_vector:
            return AsyncVectorEnv([factory for _ in range(num_envs)])
        from ..vector.sync_vec_env import SyncVectorEnv

        return SyncVectorEnv([factory for _ in range(num_envs)])
    if callable(env_id):
        env = env_id(num_envs=num_envs, seed=seed, **env_kwargs)
    else:
        if env_id not in ENV_REGISTRY:
            raise KeyError(
                f"Unknown env id '{env_id}'. Registered: {sorted(ENV_REGISTRY)}. "
                "Use register_env() for custom environments."
            )
        env = ENV_REGISTRY[env_id](num_envs=num_envs, seed=seed, **env_kwargs)
    if extra_wrappers:
        for wrapper_cls in extra_wrappers:
            env = wrapper_cls(env)
    return env
