"""Uniform optimizer wrapper.

Reference parity: ``agilerl/algorithms/core/optimizer_wrapper.py:101``.
Wraps one torch optimizer over one-or-more networks (or a ModuleDict of
per-agent networks -> one optimizer per sub-agent), supports rebuild after
architecture mutation / clone, and checkpoint round-trip.
"""

from __future__ import annotations

from typing import Any, Dict, Iterable, List, Optional, Union

import torch
import torch.nn as nn

from ...modules.base import ModuleDict

__all__ = ["OptimizerWrapper"]


class OptimizerWrapper:
    def __init__(
        self,
        optimizer_cls=None,
        networks: Optional[List[nn.Module]] = None,
        lr: float = 1e-3,
        network_names: Optional[List[str]] = None,
        lr_name: str = "lr",
        multiagent: bool = False,
        lr_critic: Optional[float] = None,
        is_llm_optimizer: bool = False,
        optimizer_kwargs: Optional[Dict[str, Any]] = None,
        **extra_kwargs: Any,
    ):
        self.optimizer_cls = optimizer_cls or torch.optim.Adam
        # reference optimizer_wrapper.py surface: optimizer_kwargs may come
        # as a dict argument (Accelerate-era calling convention) or as
        # plain **kwargs; lr_critic tags dual-lr wrappers, is_llm_optimizer
        # marks adapter-only parameter collection (both informational here)
        optimizer_kwargs = {**(optimizer_kwargs or {}), **extra_kwargs}
        self.lr_critic = lr_critic
        self.is_llm_optimizer = bool(is_llm_optimizer)
        self.lr = lr
        self.network_names = network_names or []
        self.lr_name = lr_name
        self.optimizer_kwargs = optimizer_kwargs
        self.multiagent = multiagent
        self.optimizer: Union[torch.optim.Optimizer, Dict[str, torch.optim.Optimizer], None] = None
        if networks is not None:
            self.reinit(networks, lr)

    # ------------------------------------------------------------------
    def _params(self, networks: Iterable[nn.Module]):
        params, seen = [], set()
        for net in networks:
            for p in net.parameters():
                if p.requires_grad and id(p) not in seen:
                    seen.add(id(p))
                    params.append(p)
        return params

    def reinit(self, networks: List[nn.Module], lr: Optional[float] = None) -> None:
        """(Re)build the underlying optimizer(s) over the given networks."""
        if lr is not None:
            self.lr = lr
        self._networks = list(networks)
        if self.multiagent and len(networks) >= 1 and isinstance(networks[0], ModuleDict):
            keys = list(networks[0].keys())
            self.optimizer = {}
            for k in keys:
                nets_k = [n[k] for n in networks if isinstance(n, ModuleDict) and k in n]
                self.optimizer[k] = self.optimizer_cls(
                    self._params(nets_k), lr=self.lr, **self.optimizer_kwargs
                )
        else:
            self.optimizer = self.optimizer_cls(
                self._params(networks), lr=self.lr, **self.optimizer_kwargs
            )

    # ------------------------------------------------------------------
    def zero_grad(self, set_to_none: bool = True) -> None:
        if isinstance(self.optimizer, dict):
            for opt in self.optimizer.values():
                opt.zero_grad(set_to_none=set_to_none)
        else:
            self.optimizer.zero_grad(set_to_none=set_to_none)

    def step(self) -> None:
        # Data-parallel correctness: when wrap_models attached a GradBucketer
        # (parallel/ddp.py), its async bucket allreduces must be drained and
        # scattered back BEFORE the optimizer consumes the grads — RL nets
        # rarely reach the auto-flush threshold, so without this the ranks
        # would silently diverge.
        for net in getattr(self, "_networks", []):
            bucketer = getattr(net, "_grad_bucketer", None)
            if bucketer is not None:
                bucketer.finalize()
        if isinstance(self.optimizer, dict):
            for opt in self.optimizer.values():
                opt.step()
        else:
            self.optimizer.step()

    def update_lr(self, lr: float) -> None:
        self.lr = lr
        opts = self.optimizer.values() if isinstance(self.optimizer, dict) else [self.optimizer]
        for opt in opts:
            for group in opt.param_groups:
                group["lr"] = lr

    @property
    def param_groups(self):
        if isinstance(self.optimizer, dict):
            return [g for opt in self.optimizer.values() for g in opt.param_groups]
        return self.optimizer.param_groups

    # ------------------------------------------------------------------
    def state_dict(self) -> Dict[str, Any]:
        if isinstance(self.optimizer, dict):
            return {"multiagent": {k: o.state_dict() for k, o in self.optimizer.items()}}
        return self.optimizer.state_dict()

    def load_state_dict(self, state: Dict[str, Any]) -> None:
        try:
            if isinstance(self.optimizer, dict) and "multiagent" in state:
                for k, o in self.optimizer.items():
                    if k in state["multiagent"]:
                        o.load_state_dict(state["multiagent"][k])
            elif not isinstance(self.optimizer, dict):
                self.optimizer.load_state_dict(state)
        except (ValueError, KeyError):
            # parameter shapes changed (architecture mutation) — fresh state
            pass

    def __getattr__(self, name):
        # proxy anything else to the underlying optimizer
        opt = self.__dict__.get("optimizer")
        if opt is not None and not isinstance(opt, dict):
            return getattr(opt, name)
        raise AttributeError(name)
