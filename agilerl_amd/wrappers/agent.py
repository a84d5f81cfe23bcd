"""Agent-level wrappers.

Reference parity: ``agilerl/wrappers/agent.py`` — AgentWrapper :44, RSNorm
:299 (running-statistics observation normalization wrapping any agent),
AsyncAgentsWrapper :631 (masks inactive agents in async multi-agent envs).
"""

from __future__ import annotations

from typing import Any, Dict

import numpy as np
import torch

__all__ = ["AgentWrapper", "RSNorm", "AsyncAgentsWrapper"]


class AgentWrapper:
    """Transparent proxy around an agent; subclasses intercept get_action /
    learn.  Attribute access falls through to the wrapped agent so training
    loops, tournament and mutations see the usual surface."""

    def __init__(self, agent):
        object.__setattr__(self, "agent", agent)

    def __getattr__(self, name):
        return getattr(self.agent, name)

    def __setattr__(self, name, value):
        if name in self.__dict__ or name in type(self).__dict__ or name == "agent":
            object.__setattr__(self, name, value)
        else:
            setattr(self.agent, name, value)

    def get_action(self, obs, *args, **kwargs):
        return self.agent.get_action(obs, *args, **kwargs)

    def learn(self, experiences, *args, **kwargs):
        return self.agent.learn(experiences, *args, **kwargs)

    def test(self, *args, **kwargs):
        return self.agent.test(*args, **kwargs)

    def clone(self, index=None, wrap=True):
        inner = self.agent.clone(index=index, wrap=wrap)
        clone = type(self).__new__(type(self))
        object.__setattr__(clone, "agent", inner)
        for k, v in self.__dict__.items():
            if k != "agent":
                import copy as _copy

                object.__setattr__(clone, k, _copy.deepcopy(v))
        return clone


class RunningMeanStd:
    """Welford running mean/variance over batched observations."""

    def __init__(self, shape, epsilon: float = 1e-4):
        self.mean = np.zeros(shape, dtype=np.float64)
        self.var = np.ones(shape, dtype=np.float64)
        self.count = epsilon

    def update(self, x: np.ndarray) -> None:
        x = np.asarray(x, dtype=np.float64)
        if x.ndim == len(self.mean.shape):
            x = x[None]
        batch_mean = x.mean(axis=0)
        batch_var = x.var(axis=0)
        batch_count = x.shape[0]
        delta = batch_mean - self.mean
        tot = self.count + batch_count
        self.mean = self.mean + delta * batch_count / tot
        m_a = self.var * self.count
        m_b = batch_var * batch_count
        m2 = m_a + m_b + delta**2 * self.count * batch_count / tot
        self.var = m2 / tot
        self.count = tot

    def normalize(self, x):
        return (np.asarray(x) - self.mean) / np.sqrt(self.var + 1e-8)


class RSNorm(AgentWrapper):
    """Running-statistics observation normalization around any agent.

    Observations are normalized with running mean/std both at action time
    (stats updated online during training) and inside learn() (stored
    transitions are raw; normalization applied on the way in).
    """

    def __init__(self, agent, epsilon: float = 1e-4, clip: float = 10.0,
                 norm_obs_keys=None):
        super().__init__(agent)
        space = agent.observation_space
        from ..spaces import DictSpace, TupleSpace, space_shape

        if isinstance(space, TupleSpace):
            raise TypeError("RSNorm supports flat or Dict observation spaces")
        if isinstance(space, DictSpace):
            # reference wrappers/agent.py norm_obs_keys: normalize only the
            # listed Dict-obs keys (default: every key)
            keys = list(norm_obs_keys) if norm_obs_keys is not None else list(space.spaces)
            unknown = [k for k in keys if k not in space.spaces]
            if unknown:
                raise KeyError(f"norm_obs_keys {unknown} not in observation space")
            object.__setattr__(self, "norm_obs_keys", keys)
            object.__setattr__(self, "rms", {
                k: RunningMeanStd(space_shape(space.spaces[k]), epsilon) for k in keys
            })
        else:
            object.__setattr__(self, "norm_obs_keys", None)
            object.__setattr__(self, "rms", RunningMeanStd(space_shape(space), epsilon))
        object.__setattr__(self, "clip", clip)

    def _norm(self, obs):
        if isinstance(self.rms, dict):
            out = dict(obs)
            for k in self.norm_obs_keys:
                out[k] = self._norm_one(out[k], self.rms[k])
            return out
        return self._norm_one(obs, self.rms)

    def _norm_one(self, obs, rms):
        if isinstance(obs, torch.Tensor):
            mean = torch.as_tensor(rms.mean, dtype=obs.dtype, device=obs.device)
            std = torch.as_tensor(np.sqrt(rms.var + 1e-8), dtype=obs.dtype, device=obs.device)
            return ((obs - mean) / std).clamp(-self.clip, self.clip)
        return np.clip(rms.normalize(obs), -self.clip, self.clip).astype(np.float32)

    def get_action(self, obs, *args, training: bool = True, **kwargs):
        if training:
            if isinstance(self.rms, dict):
                for k in self.norm_obs_keys:
                    v = obs[k]
                    self.rms[k].update(v if not isinstance(v, torch.Tensor) else v.cpu().numpy())
            else:
                self.rms.update(obs if not isinstance(obs, torch.Tensor) else obs.cpu().numpy())
        return self.agent.get_action(self._norm(obs), *args, training=training, **kwargs)

    def learn(self, experiences, *args, **kwargs):
        if isinstance(experiences, dict):
            experiences = dict(experiences)
            for key in ("obs", "next_obs"):
                if key not in experiences:
                    continue
                if isinstance(experiences[key], dict):
                    if isinstance(self.rms, dict):
                        experiences[key] = self._norm(experiences[key])
                else:
                    experiences[key] = self._norm(experiences[key])
        return self.agent.learn(experiences, *args, **kwargs)

    # -- persistence: normalizer stats ride inside the checkpoint --------
    def wrapper_state(self) -> dict:
        if isinstance(self.rms, dict):
            return {
                "cls": type(self).__name__,
                "keys": list(self.norm_obs_keys),
                "per_key": {k: {"mean": r.mean.copy(), "var": r.var.copy(),
                                "count": float(r.count)} for k, r in self.rms.items()},
                "clip": self.clip,
            }
        return {
            "cls": type(self).__name__,
            "mean": self.rms.mean.copy(),
            "var": self.rms.var.copy(),
            "count": float(self.rms.count),
            "clip": self.clip,
        }

    def load_wrapper_state(self, state: dict) -> None:
        if "per_key" in state:
            for k, st in state["per_key"].items():
                self.rms[k].mean[...] = st["mean"]
                self.rms[k].var[...] = st["var"]
                self.rms[k].count = st["count"]
        else:
            self.rms.mean[...] = state["mean"]
            self.rms.var[...] = state["var"]
            self.rms.count = state["count"]
        object.__setattr__(self, "clip", state.get("clip", self.clip))

    def save_checkpoint(self, path: str) -> None:
        import dill

        ckpt = self.agent.get_checkpoint_dict()
        ckpt["wrapper"] = self.wrapper_state()
        torch.save(ckpt, path, pickle_module=dill)

    def load_checkpoint(self, path: str) -> None:
        import dill

        ckpt = torch.load(path, pickle_module=dill, weights_only=False, map_location="cpu")
        self.agent._apply_checkpoint(ckpt)
        if "wrapper" in ckpt:
            self.load_wrapper_state(ckpt["wrapper"])

    def test(self, env, *args, **kwargs):
        # evaluation uses frozen stats via a normalized-view env
        wrapper_self = self

        class _NormEnv:
            def __getattr__(self, name):
                return getattr(env, name)

            def reset(self, *a, **k):
                obs, info = env.reset(*a, **k)
                return wrapper_self._norm(obs), info

            def step(self, action):
                obs, r, te, tr, info = env.step(action)
                return wrapper_self._norm(obs), r, te, tr, info

        return self.agent.test(_NormEnv(), *args, **kwargs)


class AsyncAgentsWrapper(AgentWrapper):
    """Masks out inactive sub-agents in async multi-agent envs: missing
    keys in the obs dict get placeholder zero observations and their
    actions are dropped from the returned dict."""

    def get_action(self, obs: Dict[str, Any], *args, **kwargs):
        active = [aid for aid in self.agent.agent_ids if aid in obs and obs[aid] is not None]
        full_obs = {}
        n = None
        for aid in self.agent.agent_ids:
            if aid in active:
                full_obs[aid] = obs[aid]
                n = np.asarray(obs[aid]).shape[0]
        for aid in self.agent.agent_ids:
            if aid not in active:
                from ..spaces import space_shape

                shape = space_shape(self.agent.observation_spaces[aid])
                full_obs[aid] = np.zeros((n or 1, *shape), dtype=np.float32)
        result = self.agent.get_action(full_obs, *args, **kwargs)
        if isinstance(result, tuple):
            return tuple(
                {k: v for k, v in part.items() if k in active} if isinstance(part, dict) else part
                for part in result
            )
        return {k: v for k, v in result.items() if k in active}
