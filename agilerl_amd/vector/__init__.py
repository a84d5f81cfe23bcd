from .async_vec_env import AsyncVectorEnv, AsyncPettingZooVecEnv, DummyVecEnv

__all__ = ["AsyncVectorEnv", "AsyncPettingZooVecEnv", "DummyVecEnv"]
