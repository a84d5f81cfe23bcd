from .async_vec_env import AsyncVectorEnv, AsyncPettingZooVecEnv, DummyVecEnv
from .sync_vec_env import SyncVectorEnv

__all__ = ["AsyncVectorEnv", "AsyncPettingZooVecEnv", "DummyVecEnv", "SyncVectorEnv"]
