"""Reference-checkpoint cross-compatibility.

The reference framework writes classic-agent checkpoints as ONE dill-
pickled dict (``agilerl/algorithms/core/base.py:1128-1138``) whose layout
is (``get_checkpoint_dict``, ``base.py:315-372``):

- every inspectable attribute at the TOP level (lr, batch_size, fitness,
  steps, ...), plus ``agilerl_version``;
- ``network_info``: ``{"modules": {f"{name}_cls", f"{name}_init_dict",
  f"{name}_state_dict", f"{name}_module_dict_cls"}, "optimizers": {...},
  "network_names": [...], "optimizer_names": [...]}``.

This module makes :meth:`EvolvableAlgorithm.load_checkpoint` accept that
layout.  Two translation layers:

1. **Unpickling**: reference pickles contain class objects from
   ``agilerl.*`` and ``gymnasium.spaces.*`` — neither is importable here.
   :class:`RefCompatUnpickler` maps them BY CLASS NAME onto this
   package's equivalents (QNetwork -> agilerl_amd QNetwork, gymnasium Box
   -> agilerl_amd Box, ...).  Unknown classes raise with the offending
   dotted path so failures are informative rather than silent.
2. **Layout**: :func:`convert_reference_checkpoint` reshapes the flat
   reference dict into this package's nested checkpoint format, filtering
   each init dict to the kwargs our constructors accept (the reference
   does the same on load — ``filter_init_dict``).

Deliberate divergences (documented in docs/checkpoints.md): optimizer
moment tensors transfer only when parameter shapes line up; reference
``net_config``-era fields our constructors don't take are dropped with a
recorded list in ``ckpt["_ref_compat_dropped"]``.
"""

from __future__ import annotations

import importlib
from typing import Any, Dict, List, Tuple

import numpy as np
import torch

__all__ = [
    "RefCompatUnpickler",
    "is_reference_layout",
    "convert_reference_checkpoint",
    "load_checkpoint_file",
]

# class-name -> (module path here).  Covers the evolvable module /
# network / algorithm surface the reference stores in `{name}_cls`.
_NAME_MAP: Dict[str, str] = {
    # networks
    "QNetwork": "agilerl_amd.networks.q_networks",
    "RainbowQNetwork": "agilerl_amd.networks.q_networks",
    "ContinuousQNetwork": "agilerl_amd.networks.q_networks",
    "ValueNetwork": "agilerl_amd.networks.value_networks",
    "DeterministicActor": "agilerl_amd.networks.actors",
    "StochasticActor": "agilerl_amd.networks.actors",
    # modules
    "EvolvableMLP": "agilerl_amd.modules.mlp",
    "EvolvableCNN": "agilerl_amd.modules.cnn",
    "EvolvableLSTM": "agilerl_amd.modules.lstm",
    "EvolvableMultiInput": "agilerl_amd.modules.multi_input",
    "EvolvableSimBa": "agilerl_amd.modules.simba",
    "EvolvableResNet": "agilerl_amd.modules.resnet",
    "EvolvableGPT": "agilerl_amd.modules.gpt",
    "EvolvableBERT": "agilerl_amd.modules.bert",
    "ModuleDict": "agilerl_amd.modules.base",
    "NoisyLinear": "agilerl_amd.modules.custom_components",
    # algorithms (for `algo_cls` if present)
    "DQN": "agilerl_amd.algorithms.dqn",
    "RainbowDQN": "agilerl_amd.algorithms.dqn_rainbow",
    "CQN": "agilerl_amd.algorithms.cqn",
    "DDPG": "agilerl_amd.algorithms.ddpg",
    "TD3": "agilerl_amd.algorithms.td3",
    "PPO": "agilerl_amd.algorithms.ppo",
    "MADDPG": "agilerl_amd.algorithms.maddpg",
    "MATD3": "agilerl_amd.algorithms.matd3",
    "IPPO": "agilerl_amd.algorithms.ippo",
    # registry / config carriers
    "HyperparameterConfig": "agilerl_amd.algorithms.core.registry",
    "RLParameter": "agilerl_amd.algorithms.core.registry",
    "NetworkGroup": "agilerl_amd.algorithms.core.registry",
    # spaces (gymnasium)
    "Box": "agilerl_amd.spaces",
    "Discrete": "agilerl_amd.spaces",
    "MultiDiscrete": "agilerl_amd.spaces",
    "MultiBinary": "agilerl_amd.spaces",
}
# gymnasium composite spaces have name clashes with builtins
_GYM_SPECIAL = {"Dict": "DictSpace", "Tuple": "TupleSpace"}


def _resolve(name: str, ref_module: str):
    if ref_module.startswith("gymnasium") and name in _GYM_SPECIAL:
        mod = importlib.import_module("agilerl_amd.spaces")
        return getattr(mod, _GYM_SPECIAL[name])
    target = _NAME_MAP.get(name)
    if target is None:
        raise ModuleNotFoundError(
            f"reference checkpoint references {ref_module}.{name}, which has "
            f"no mapped equivalent in agilerl_amd (add it to "
            f"utils/ref_compat._NAME_MAP if it should load)"
        )
    return getattr(importlib.import_module(target), name)


def _gym_space_to_ours(obj: Any) -> Any:
    """Translate an unpickled gymnasium-space state dict onto our spaces.

    Gym spaces restore via ``__dict__`` (no __init__ call), so after the
    class swap the instance has gym's attribute names (``_shape``,
    ``bounded_below``...).  Normalize the ones our code reads."""
    d = getattr(obj, "__dict__", None)
    if d is None:
        return obj
    if "_shape" in d and "shape" not in d:
        obj.shape = tuple(d["_shape"])
    if "n" in d and hasattr(obj, "n"):
        try:
            obj.n = int(obj.n)
        except (TypeError, ValueError):
            pass
    if not hasattr(obj, "_rng"):
        obj._rng = np.random.default_rng()
    if "dtype" in d and not isinstance(obj.dtype, np.dtype):
        try:
            obj.dtype = np.dtype(obj.dtype)
        except TypeError:
            obj.dtype = np.dtype(np.float32)
    return obj


def _make_unpickler(base_unpickler):
    class RefCompatUnpickler(base_unpickler):  # type: ignore[misc, valid-type]
        def find_class(self, module: str, name: str):
            if module.startswith("agilerl.") or module == "agilerl":
                return _resolve(name, module)
            if module.startswith("gymnasium"):
                return _resolve(name, module)
            return super().find_class(module, name)

    return RefCompatUnpickler


def RefCompatUnpickler(file):  # noqa: N802 - factory with class-like name
    import dill

    return _make_unpickler(dill.Unpickler)(file)


def load_checkpoint_file(path: str, device: str = "cpu") -> Dict[str, Any]:
    """torch.load with the class-mapping unpickler (handles both this
    package's checkpoints and reference-written ones)."""
    import dill

    class _Mod:
        Unpickler = _make_unpickler(dill.Unpickler)
        # torch.load uses pickle_module.Unpickler and .load attributes
        load = staticmethod(dill.load)

    ckpt = torch.load(
        path, map_location="cpu", pickle_module=_Mod, weights_only=False
    )
    return ckpt


def is_reference_layout(ckpt: Dict[str, Any]) -> bool:
    ni = ckpt.get("network_info")
    return isinstance(ni, dict) and "network_names" in ni and "attributes" not in ckpt


def _filter_init_dict(cls, init: Dict[str, Any]) -> Tuple[Dict[str, Any], List[str]]:
    import inspect

    # union of NAMED parameters across the MRO: classes taking **kwargs
    # forward them upward, so the accepted surface is every named param of
    # every ancestor __init__ — unknown reference fields still drop instead
    # of reaching a strict base constructor
    named = set()
    for klass in cls.__mro__:
        init_fn = klass.__dict__.get("__init__")
        if init_fn is None:
            continue
        try:
            for p in inspect.signature(init_fn).parameters.values():
                if p.kind in (inspect.Parameter.POSITIONAL_OR_KEYWORD,
                              inspect.Parameter.KEYWORD_ONLY):
                    named.add(p.name)
        except (TypeError, ValueError):
            continue
    named.discard("self")
    kept, dropped = {}, []
    for k, v in init.items():
        if k in named:
            kept[k] = _normalize_value(v)
        else:
            dropped.append(k)
    return kept, dropped


def _normalize_value(v: Any) -> Any:
    # gym-space shims restored by the unpickler
    from ..spaces import Space

    if isinstance(v, Space):
        return _gym_space_to_ours(v)
    return v


def convert_reference_checkpoint(ckpt: Dict[str, Any], device: str = "cpu") -> Dict[str, Any]:
    """Reference flat layout -> this package's nested checkpoint dict."""
    ni = ckpt["network_info"]
    ref_modules: Dict[str, Any] = ni.get("modules", {})
    dropped_all: List[str] = []
    modules: Dict[str, Any] = {}
    for name in ni.get("network_names", []):
        cls = ref_modules.get(f"{name}_cls")
        if cls is None:
            continue
        init = dict(ref_modules.get(f"{name}_init_dict", {}) or {})
        state = ref_modules.get(f"{name}_state_dict", {}) or {}
        if isinstance(cls, dict):  # multi-agent ModuleDict entry
            from ..modules.base import ModuleDict

            sub = {}
            for agent_id, sub_cls in cls.items():
                sub_init, dropped = _filter_init_dict(sub_cls, init.get(agent_id, {}))
                dropped_all += [f"{name}.{agent_id}.{d}" for d in dropped]
                sub[agent_id] = {
                    "module_cls": sub_cls,
                    "init_dict": sub_init,
                    "state_dict": state.get(agent_id, {}),
                }
            modules[name] = {"module_dict": sub, "module_dict_cls": ModuleDict}
        else:
            kept, dropped = _filter_init_dict(cls, init)
            dropped_all += [f"{name}.{d}" for d in dropped]
            modules[name] = {
                "module_cls": cls,
                "init_dict": kept,
                "state_dict": state,
            }
    optimizers = {}
    for opt_name in ni.get("optimizer_names", []):
        blob = ni.get("optimizers", {})
        # reference OptimizerWrapper.checkpoint_dict stores
        # f"{name}_state_dict" (plus metadata we rebuild from the registry)
        state = blob.get(f"{opt_name}_state_dict", blob.get(opt_name))
        if state is not None:
            optimizers[opt_name] = state

    reserved = {"network_info", "agilerl_version", "accelerator"}
    attributes = {
        k: _normalize_value(v) for k, v in ckpt.items() if k not in reserved
    }
    return {
        "agilerl_version": ckpt.get("agilerl_version", "reference"),
        "algo": attributes.get("algo"),
        "attributes": attributes,
        "network_info": {"modules": modules, "optimizers": optimizers},
        "_ref_compat_dropped": dropped_all,
    }
