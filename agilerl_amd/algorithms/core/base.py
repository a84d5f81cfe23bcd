"""Algorithm base classes: the evolvable-agent object model.

Reference parity: ``agilerl/algorithms/core/base.py`` (EvolvableAlgorithm
:393, RegistryMeta :295, RLAlgorithm :1461, MultiAgentRLAlgorithm :1580;
clone :1060, checkpoint :1128-1273).  Re-designed:

- ``AlgorithmMeta`` captures constructor args on the most-derived
  ``__init__`` call and runs registry sanity checks after construction.
- Networks are registered in :class:`NetworkGroup`-s; optimizers in
  :class:`OptimizerConfig`-s.  ``clone()`` = rebuild from captured init
  args (with live mutable-hyperparameter values) + ``module.clone()`` for
  every registered network (which reproduces mutated architectures
  exactly) + optimizer rebuild.
- Checkpoints are a single ``.pt`` dict (saved with dill) holding init
  args, per-module ``init_dict``/``state_dict`` (``network_info``),
  optimizer state and tracked attributes — mirroring the reference's
  on-disk layout.
"""

from __future__ import annotations

import inspect
from typing import Any, Dict, List, Optional, Tuple, Type

import numpy as np
import torch
import torch.nn as nn

from ...modules.base import (
    EvolvableModule,
    ModuleDict,
    load_module_from_checkpoint,
    module_checkpoint_dict,
)
from ...spaces import Space
from .optimizer_wrapper import OptimizerWrapper
from .registry import HyperparameterConfig, MutationRegistry, NetworkGroup, OptimizerConfig

__all__ = ["EvolvableAlgorithm", "RLAlgorithm", "MultiAgentRLAlgorithm", "AlgorithmMeta"]

AGILERL_AMD_VERSION = "0.1.0"


class AlgorithmMeta(type):
    """Captures init args of the most-derived constructor call."""

    def __new__(mcs, name, bases, namespace):
        cls = super().__new__(mcs, name, bases, namespace)
        init = namespace.get("__init__")
        if init is not None:
            sig = inspect.signature(init)

            def wrapped_init(self, *args, __init=init, __sig=sig, **kwargs):
                outermost = not hasattr(self, "_init_args")
                if outermost:
                    try:
                        bound = __sig.bind(self, *args, **kwargs)
                        bound.apply_defaults()
                        captured = {
                            k: v
                            for k, v in list(bound.arguments.items())[1:]
                            if k not in ("args", "kwargs")
                        }
                        if "kwargs" in bound.arguments:
                            captured.update(bound.arguments["kwargs"])
                        object.__setattr__(self, "_init_args", captured)
                    except TypeError:
                        object.__setattr__(self, "_init_args", {})
                __init(self, *args, **kwargs)
                if outermost:
                    post = getattr(self, "_registry_init", None)
                    if post is not None:
                        post()

            wrapped_init.__signature__ = sig
            wrapped_init.__doc__ = init.__doc__
            cls.__init__ = wrapped_init
        return cls


class EvolvableAlgorithm(metaclass=AlgorithmMeta):
    """Base class for every agent in a population."""

    def __init__(
        self,
        index: int = 0,
        learn_step: int = 1,
        device: str = "cpu",
        hp_config: Optional[HyperparameterConfig] = None,
        name: Optional[str] = None,
    ):
        self.index = index
        self.learn_step = learn_step
        self.device = device
        self.hp_config = hp_config or HyperparameterConfig()
        self.algo = name or type(self).__name__
        self.registry = MutationRegistry()

        self.fitness: List[float] = []
        self.scores: List[float] = []
        self.steps: List[int] = [0]
        self.mut: str = "None"
        self._wrapped: Dict[str, nn.Module] = {}

    # ------------------------------------------------------------------
    # Reference-API compatibility kwargs
    # ------------------------------------------------------------------
    _COMPAT_KWARGS = ("accelerator", "wrap", "mut", "normalize_images", "torch_compiler")

    def _accept_compat_kwargs(self, **kw) -> None:
        """Accept the reference constructor kwargs that have no MI355X-native
        role (reference base.py:819-845 Accelerate wrap, torch.compile).

        - ``accelerator``/``wrap``: HF-Accelerate integration — replaced here
          by one-process-per-GPU RCCL (``agilerl_amd.parallel``); warned if set.
        - ``torch_compiler``: hot paths run hand-written HIP kernels and
          hipGraph capture instead of a tracing compiler; warned if set.
        - ``mut``: last-mutation tag (stored; the HPO loop overwrites it).
        - ``normalize_images``: uint8 image obs are always scaled by 1/255
          (warned when explicitly disabled).
        Unknown keys raise TypeError exactly like a plain signature would.
        """
        import warnings

        unknown = set(kw) - set(self._COMPAT_KWARGS)
        if unknown:
            raise TypeError(
                f"{type(self).__name__}.__init__() got an unexpected keyword "
                f"argument '{sorted(unknown)[0]}'"
            )
        if kw.get("accelerator") is not None:
            warnings.warn(
                "`accelerator` is accepted for reference-API compatibility but "
                "ignored: use agilerl_amd.parallel (one process per GPU over "
                "RCCL) for distributed training.",
                RuntimeWarning,
            )
            if hasattr(self, "_init_args"):
                self._init_args["accelerator"] = None  # keep clones picklable
        self.accelerator = None
        if kw.get("torch_compiler"):
            warnings.warn(
                "`torch_compiler` ignored: hot paths use hand-written HIP "
                "kernels and hipGraph capture.",
                RuntimeWarning,
            )
        if kw.get("mut") is not None:
            self.mut = kw["mut"]
        self.normalize_images = bool(kw.get("normalize_images", True))
        if not self.normalize_images:
            warnings.warn(
                "normalize_images=False ignored: uint8 image observations are "
                "always scaled by 1/255.",
                RuntimeWarning,
            )

    # ------------------------------------------------------------------
    # Registry
    # ------------------------------------------------------------------
    def register_network_group(self, group: NetworkGroup) -> None:
        self.registry.register_group(group)

    def register_optimizer(self, config: OptimizerConfig) -> None:
        self.registry.register_optimizer(config)

    def register_mutation_hook(self, method_name: str) -> None:
        self.registry.register_hook(method_name)

    def _registry_init(self) -> None:
        """Post-construction invariants (reference base.py:750-817)."""
        if not self.registry.groups:
            raise RuntimeError(
                f"{type(self).__name__} registered no NetworkGroup — every algorithm "
                "must call register_network_group() in __init__."
            )
        n_policies = sum(1 for g in self.registry.groups if g.policy)
        if n_policies != 1:
            raise RuntimeError(
                f"{type(self).__name__} must register exactly one policy NetworkGroup "
                f"(got {n_policies})."
            )
        for group in self.registry.groups:
            for attr in group.all_names():
                if not hasattr(self, attr):
                    raise RuntimeError(
                        f"NetworkGroup references missing attribute '{attr}' on {type(self).__name__}."
                    )
        for cfg in self.registry.optimizer_configs:
            if not hasattr(self, cfg.name):
                raise RuntimeError(
                    f"OptimizerConfig references missing attribute '{cfg.name}'."
                )

    # ------------------------------------------------------------------
    # Network / optimizer access
    # ------------------------------------------------------------------
    def evolvable_networks(self) -> Dict[str, EvolvableModule]:
        return {name: getattr(self, name) for name in self.registry.all_network_names()}

    @property
    def policy_network(self) -> EvolvableModule:
        return getattr(self, self.registry.policy_group.eval_network)

    def _reinit_optimizers(self) -> None:
        for cfg in self.registry.optimizer_configs:
            wrapper: OptimizerWrapper = getattr(self, cfg.name)
            nets = [getattr(self, n) for n in cfg.networks]
            lr = getattr(self, cfg.lr_name, wrapper.lr)
            wrapper.reinit(nets, lr=lr)

    def mutation_hook(self) -> None:
        """Called after any architecture/parameter mutation."""
        self._reinit_optimizers()
        for hook_name in self.registry.hooks:
            getattr(self, hook_name)()

    # ------------------------------------------------------------------
    # Architecture mutation surface (driven by hpo.Mutations)
    # ------------------------------------------------------------------
    def apply_architecture_mutation(self, method: str, **choices) -> dict:
        """Apply ``method`` to every registered network that supports it.

        The policy eval network samples the random choices; the identical
        mutation is replayed on every other registered network (targets,
        critics) so group structure stays consistent.
        """
        policy = self.policy_network
        result = policy.apply_mutation(method, **choices) or {}
        merged = {**choices, **result}
        policy_enc = getattr(policy, "encoder", None)
        for name, net in self.evolvable_networks().items():
            if net is policy:
                continue
            shares_encoder = (
                policy_enc is not None and getattr(net, "encoder", None) is policy_enc
            )
            if shares_encoder and method.startswith("encoder."):
                continue  # the shared encoder object was already mutated once
            if shares_encoder and method in ("add_latent_node", "remove_latent_node"):
                net._resize_latent(policy.latent_dim, resize_encoder=False)
                continue
            if method in net.mutation_methods:
                net.apply_mutation(method, **merged)
        self.mutation_hook()
        return merged

    @property
    def mutation_methods(self) -> List[str]:
        return self.policy_network.mutation_methods

    # ------------------------------------------------------------------
    # Tracking
    # ------------------------------------------------------------------
    @property
    def fitness_score(self) -> float:
        return self.fitness[-1] if self.fitness else -np.inf

    # ------------------------------------------------------------------
    # Cloning (evolution)
    # ------------------------------------------------------------------
    # Never cloned/checkpointed: captured hipGraphs are bound to ONE agent's
    # live tensors — a copied graph handle replayed by an offspring reads
    # freed/foreign memory (observed as an HSA hardware exception).
    _GRAPH_ATTRS = ("_graph", "_graph_static", "_learn_graph", "_learn_static")

    def inspect_attributes(self, ignore: Tuple[str, ...] = ()) -> Dict[str, Any]:
        """Plain (non-network, non-optimizer) attributes to carry across clones."""
        skip = set(ignore) | {
            "registry",
            "hp_config",
            "_init_args",
            "_wrapped",
            "accelerator",
            "_decode_engine",  # paged-KV pools + model ref; rebuilt lazily
        }
        skip.update(self._GRAPH_ATTRS)
        skip.update(self.registry.all_network_names())
        skip.update(cfg.name for cfg in self.registry.optimizer_configs)
        out = {}
        for k, v in vars(self).items():
            if k in skip or isinstance(v, (nn.Module, OptimizerWrapper)):
                continue
            if type(v).__name__ == "CUDAGraph":
                continue
            out[k] = v
        return out

    def _live_init_args(self) -> Dict[str, Any]:
        """Captured init args, refreshed with live mutable-HP values."""
        init = dict(getattr(self, "_init_args", {}))
        for k in list(init.keys()):
            if hasattr(self, k) and not isinstance(
                getattr(self, k), (nn.Module, OptimizerWrapper)
            ):
                init[k] = getattr(self, k)
        init["hp_config"] = self.hp_config
        return init

    def clone(self, index: Optional[int] = None, wrap: bool = True) -> "EvolvableAlgorithm":
        clone = type(self)(**self._live_init_args())
        # exact architecture + weights for every registered network
        for name, net in self.evolvable_networks().items():
            setattr(clone, name, net.clone())
        clone._reinit_optimizers()
        for cfg in self.registry.optimizer_configs:
            getattr(clone, cfg.name).load_state_dict(getattr(self, cfg.name).state_dict())
        # tracked / mutable plain attributes
        for k, v in self.inspect_attributes().items():
            try:
                import copy as _copy

                setattr(clone, k, _copy.deepcopy(v))
            except Exception:
                setattr(clone, k, v)
        clone.index = self.index if index is None else index
        clone.post_clone_hook(self)
        return clone

    def post_clone_hook(self, parent: "EvolvableAlgorithm") -> None:
        """Subclass hook (e.g. re-point derived references after clone)."""

    # ------------------------------------------------------------------
    # Checkpointing (SURVEY §2.6 format)
    # ------------------------------------------------------------------
    def get_checkpoint_dict(self) -> Dict[str, Any]:
        return {
            "agilerl_version": AGILERL_AMD_VERSION,
            "algo": self.algo,
            "algo_cls": type(self),
            "init_args": self._serializable_init_args(),
            "attributes": self.inspect_attributes(),
            "network_info": {
                "modules": {
                    name: module_checkpoint_dict(net)
                    for name, net in self.evolvable_networks().items()
                },
                "optimizers": {
                    cfg.name: getattr(self, cfg.name).state_dict()
                    for cfg in self.registry.optimizer_configs
                },
            },
        }

    def _serializable_init_args(self) -> Dict[str, Any]:
        args = self._live_init_args()
        args.pop("accelerator", None)
        return args

    def save_checkpoint(self, path: str) -> None:
        import dill

        torch.save(self.get_checkpoint_dict(), path, pickle_module=dill)

    def load_checkpoint(self, path: str) -> None:
        """Load this package's OR a reference-written checkpoint file.

        Reference format (``agilerl/algorithms/core/base.py:315-372``):
        flat attribute dict + ``network_info`` with ``{name}_cls/
        {name}_init_dict/{name}_state_dict`` keys, pickled classes from
        ``agilerl.*``/``gymnasium.*`` — translated by utils/ref_compat.
        """
        from ...utils.ref_compat import (
            convert_reference_checkpoint,
            is_reference_layout,
            load_checkpoint_file,
        )

        ckpt = load_checkpoint_file(path)
        if is_reference_layout(ckpt):
            ckpt = convert_reference_checkpoint(ckpt, device=self.device)
        self._apply_checkpoint(ckpt)

    def _apply_checkpoint(self, ckpt: Dict[str, Any]) -> None:
        for name, mod_ckpt in ckpt["network_info"]["modules"].items():
            setattr(self, name, load_module_from_checkpoint(mod_ckpt, device=self.device))
        self._reinit_optimizers()
        for opt_name, opt_state in ckpt["network_info"]["optimizers"].items():
            if hasattr(self, opt_name):
                getattr(self, opt_name).load_state_dict(opt_state)
        for k, v in ckpt["attributes"].items():
            if k == "device" or k.startswith("_ref_compat"):
                continue  # receiving agent keeps its own device
            setattr(self, k, v)

    @classmethod
    def load(cls, path: str, device: str = "cpu") -> "EvolvableAlgorithm":
        from ...utils.ref_compat import (
            convert_reference_checkpoint,
            is_reference_layout,
            load_checkpoint_file,
        )

        ckpt = load_checkpoint_file(path)
        if is_reference_layout(ckpt):
            ckpt = convert_reference_checkpoint(ckpt, device=device)
            algo_cls = ckpt.get("attributes", {}).get("algo_cls")
            if algo_cls is None:
                algo_name = ckpt.get("attributes", {}).get("algo")
                if isinstance(algo_name, str):
                    from ...utils.ref_compat import _NAME_MAP
                    import importlib as _il

                    mod = _NAME_MAP.get(algo_name)
                    algo_cls = getattr(_il.import_module(mod), algo_name) if mod else cls
                else:
                    algo_cls = cls
            # reference checkpoints carry constructor kwargs as flat
            # attributes; filter against the target constructor
            import inspect as _inspect

            params = _inspect.signature(algo_cls.__init__).parameters
            init_args = {
                k: v for k, v in ckpt["attributes"].items()
                if k in params and k != "self"
            }
        else:
            algo_cls = ckpt.get("algo_cls", cls)
            init_args = dict(ckpt["init_args"])
        init_args["device"] = device
        agent = algo_cls(**init_args)
        agent._apply_checkpoint(ckpt)
        wrapper = ckpt.get("wrapper")
        if wrapper and wrapper.get("cls") == "RSNorm":
            from ...wrappers.agent import RSNorm

            wrapped = RSNorm(agent)
            wrapped.load_wrapper_state(wrapper)
            return wrapped
        return agent

    # ------------------------------------------------------------------
    # Device / distributed plumbing
    # ------------------------------------------------------------------
    def to_device(self, device: str) -> "EvolvableAlgorithm":
        self.device = device
        for name, net in self.evolvable_networks().items():
            net = net.to(device)
            # every evolvable sub-module rebuilds onto ITS OWN device attr
            # during mutations — refresh them all or post-move mutations
            # would recreate layers on the old device
            for mod in net.modules():
                if hasattr(mod, "device"):
                    mod.device = device
            setattr(self, name, net)
        self._reinit_optimizers()
        return self

    def wrap_models(self, process_group=None) -> None:
        """Wrap eval networks for data-parallel training (RCCL DDP).

        Uses the first-party bucketed-allreduce wrapper in
        ``agilerl_amd.parallel`` (no Accelerate/DeepSpeed).
        """
        from ...parallel import DistributedState, wrap_ddp

        state = DistributedState.get()
        if not state.is_distributed:
            return
        for name in self.registry.eval_network_names():
            net = getattr(self, name)
            wrap_ddp(net, process_group)

    def unwrap_models(self) -> None:
        pass  # wrap_ddp hooks gradients in place; nothing to unwrap

    # ------------------------------------------------------------------
    # Abstract API
    # ------------------------------------------------------------------
    def get_action(self, obs, **kwargs):  # pragma: no cover - interface
        raise NotImplementedError

    def learn(self, experiences, **kwargs):  # pragma: no cover - interface
        raise NotImplementedError

    def test(self, env, max_steps=None, loop=3, **kwargs):  # pragma: no cover
        raise NotImplementedError

    def recompile(self, mode: str = "default") -> None:
        """torch.compile the registered networks (hipGraph-friendly)."""
        for name in self.registry.eval_network_names():
            try:
                setattr(self, name, torch.compile(getattr(self, name), mode=mode))
            except Exception:
                pass

    # convenience
    def obs_to_tensor(self, obs) -> Any:
        return obs_to_device(obs, self.device)


def obs_to_device(obs, device):
    if isinstance(obs, dict):
        return {k: obs_to_device(v, device) for k, v in obs.items()}
    if isinstance(obs, (tuple, list)):
        return type(obs)(obs_to_device(v, device) for v in obs)
    if isinstance(obs, torch.Tensor):
        return obs.to(device)
    return torch.as_tensor(np.asarray(obs)).to(device)


class RLAlgorithm(EvolvableAlgorithm):
    """Single-agent RL base (reference base.py:1461)."""

    def __init__(
        self,
        observation_space: Space,
        action_space: Space,
        index: int = 0,
        learn_step: int = 1,
        device: str = "cpu",
        hp_config: Optional[HyperparameterConfig] = None,
        name: Optional[str] = None,
    ):
        super().__init__(index=index, learn_step=learn_step, device=device, hp_config=hp_config, name=name)
        self.observation_space = observation_space
        self.action_space = action_space

    # ------------------------------------------------------------------
    @classmethod
    def population(
        cls,
        size: int,
        observation_space: Space,
        action_space: Space,
        device: str = "cpu",
        **kwargs,
    ) -> List["RLAlgorithm"]:
        """Build a population of agents with indices 0..size-1."""
        return [
            cls(observation_space, action_space, index=i, device=device, **kwargs)
            for i in range(size)
        ]

    # Generic vec-env evaluation: mean episodic return over `loop` episodes/env.
    def test(self, env, max_steps: Optional[int] = None, loop: int = 3, swap_channels: bool = False) -> float:
        with torch.no_grad():
            rewards = []
            for _ in range(loop):
                obs, _ = env.reset()
                done_mask = np.zeros(env.num_envs, dtype=bool)
                ep_rew = np.zeros(env.num_envs, dtype=np.float64)
                steps = 0
                while not done_mask.all():
                    action = self.get_action(obs, training=False)
                    obs, rew, term, trunc, _ = env.step(action)
                    rew = np.asarray(rew, dtype=np.float64)
                    ep_rew += rew * (~done_mask)
                    done_mask |= np.asarray(term) | np.asarray(trunc)
                    steps += 1
                    if max_steps is not None and steps >= max_steps:
                        break
                rewards.append(ep_rew.mean())
        fitness = float(np.mean(rewards))
        self.fitness.append(fitness)
        return fitness


class MultiAgentRLAlgorithm(EvolvableAlgorithm):
    """Multi-agent RL base (reference base.py:1580).

    Networks are :class:`ModuleDict`-s keyed by agent id; observation /
    action spaces are dicts keyed the same way.
    """

    def __init__(
        self,
        observation_spaces: Dict[str, Space],
        action_spaces: Dict[str, Space],
        agent_ids: Optional[List[str]] = None,
        index: int = 0,
        learn_step: int = 1,
        device: str = "cpu",
        hp_config: Optional[HyperparameterConfig] = None,
        name: Optional[str] = None,
    ):
        super().__init__(index=index, learn_step=learn_step, device=device, hp_config=hp_config, name=name)
        self.observation_spaces = dict(observation_spaces)
        self.action_spaces = dict(action_spaces)
        self.agent_ids = list(agent_ids) if agent_ids is not None else list(observation_spaces.keys())
        self.n_agents = len(self.agent_ids)

    @classmethod
    def population(
        cls,
        size: int,
        observation_spaces: Dict[str, Space],
        action_spaces: Dict[str, Space],
        device: str = "cpu",
        **kwargs,
    ) -> List["MultiAgentRLAlgorithm"]:
        return [
            cls(observation_spaces, action_spaces, index=i, device=device, **kwargs)
            for i in range(size)
        ]
