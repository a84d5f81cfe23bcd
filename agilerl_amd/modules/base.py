"""Evolvable module object model.

This is the foundation of the evolutionary-HPO object model (reference:
``agilerl/modules/base.py`` — ``EvolvableModule`` :271, ``ModuleMeta`` :239,
``@mutation`` :30). Re-designed from scratch for this framework:

- Every evolvable building block subclasses :class:`EvolvableModule`.
- Constructor arguments are captured automatically (via ``ModuleMeta``)
  into ``init_dict`` so modules can be re-built (clone / checkpoint /
  architecture mutation with parameter preservation).
- Architecture-mutation methods are declared with the :func:`mutation`
  decorator.  A mutation method *returns a dict of the random choices it
  made* (or ``None``); the wrapper records ``(method_name, choices)`` in
  ``last_mutation`` so the identical mutation can be replayed on the other
  networks of the same group (e.g. the target network) via
  :meth:`EvolvableModule.apply_mutation`.
- Parameter preservation across rebuilds is :func:`preserve_parameters`:
  overlapping slices of every matching parameter/buffer are copied.
"""

from __future__ import annotations

import copy
import inspect
from enum import Enum
from typing import Any, Callable, Dict, List, Optional, Tuple

import torch
import torch.nn as nn

__all__ = [
    "MutationType",
    "mutation",
    "EvolvableModule",
    "EvolvableWrapper",
    "ModuleDict",
    "preserve_parameters",
    "module_checkpoint_dict",
    "load_module_from_checkpoint",
]


class MutationType(str, Enum):
    LAYER = "layer"
    NODE = "node"
    ACTIVATION = "activation"


def mutation(mut_type: MutationType) -> Callable:
    """Decorator tagging a method as an architecture mutation.

    The wrapped method should perform the mutation (usually ending in a
    ``recreate_network()`` call) and return a dict of any *sampled* choices
    so the mutation can be replayed deterministically on sibling networks.
    """

    def decorator(fn: Callable) -> Callable:
        fn._mutation_type = mut_type
        return fn

    return decorator


def _wrap_mutation_method(name: str, fn: Callable) -> Callable:
    def wrapper(self, *args, **kwargs):
        result = fn(self, *args, **kwargs)
        # Record on the *outermost* module the call was made on.
        self._last_mutation = (name, result if isinstance(result, dict) else {})
        return result

    wrapper.__name__ = fn.__name__
    wrapper.__doc__ = fn.__doc__
    wrapper.__signature__ = inspect.signature(fn)
    wrapper._mutation_type = fn._mutation_type
    wrapper._mutation_wrapped = True
    return wrapper


class ModuleMeta(type):
    """Metaclass: captures ``__init__`` args and wraps mutation methods."""

    def __new__(mcs, name, bases, namespace):
        for attr, val in list(namespace.items()):
            if callable(val) and getattr(val, "_mutation_type", None) is not None and not getattr(
                val, "_mutation_wrapped", False
            ):
                namespace[attr] = _wrap_mutation_method(attr, val)
        cls = super().__new__(mcs, name, bases, namespace)
        init = namespace.get("__init__")
        if init is not None:
            sig = inspect.signature(init)

            def wrapped_init(self, *args, __init=init, __sig=sig, **kwargs):
                if not hasattr(self, "_init_args"):
                    try:
                        bound = __sig.bind(self, *args, **kwargs)
                        bound.apply_defaults()
                        var_kw = {
                            p.name for p in __sig.parameters.values()
                            if p.kind is inspect.Parameter.VAR_KEYWORD
                        }
                        captured = {
                            k: v
                            for k, v in list(bound.arguments.items())[1:]
                            if k != "args" and k not in var_kw
                        }
                        # flatten the var-keyword dict (whatever it's named)
                        # so clones re-pass its contents as real kwargs
                        for k in var_kw:
                            captured.update(bound.arguments.get(k, {}))
                        object.__setattr__(self, "_init_args", captured)
                    except TypeError:
                        object.__setattr__(self, "_init_args", {})
                return __init(self, *args, **kwargs)

            wrapped_init.__signature__ = sig
            wrapped_init.__doc__ = init.__doc__
            cls.__init__ = wrapped_init
        return cls


class EvolvableModule(nn.Module, metaclass=ModuleMeta):
    """Base class for all evolvable building blocks."""

    def __init__(self, device: str = "cpu", name: Optional[str] = None,
                 random_seed: Optional[int] = None):
        super().__init__()
        self.device = device
        # reference modules/base.py: `name` labels the module in mutation
        # logs; `random_seed` makes weight init reproducible
        self.name = name or type(self).__name__.lower()
        self.random_seed = random_seed
        if random_seed is not None:
            torch.manual_seed(int(random_seed))
        self._last_mutation: Optional[Tuple[str, dict]] = None

    # ------------------------------------------------------------------
    # Introspection
    # ------------------------------------------------------------------
    @property
    def init_dict(self) -> Dict[str, Any]:
        """Constructor arguments this module was built with (deep-copied)."""
        out = {}
        for k, v in getattr(self, "_init_args", {}).items():
            # Live (possibly mutated) values take precedence when the module
            # exposes an attribute of the same name.
            live = getattr(self, k, v)
            if isinstance(live, (nn.Module, torch.Tensor)):
                live = v
            try:
                out[k] = copy.deepcopy(live)
            except Exception:
                out[k] = live
        return out

    @classmethod
    def mutation_method_names(cls) -> List[str]:
        names = []
        for attr in dir(cls):
            val = getattr(cls, attr, None)
            if callable(val) and getattr(val, "_mutation_type", None) is not None:
                names.append(attr)
        return sorted(names)

    @property
    def mutation_methods(self) -> List[str]:
        return type(self).mutation_method_names()

    def get_mutation_methods(self) -> Dict[str, MutationType]:
        return {n: getattr(type(self), n)._mutation_type for n in self.mutation_methods}

    @property
    def last_mutation(self) -> Optional[Tuple[str, dict]]:
        return self._last_mutation

    @property
    def last_mutation_attr(self) -> Optional[str]:
        return self._last_mutation[0] if self._last_mutation else None

    # ------------------------------------------------------------------
    # Mutation replay
    # ------------------------------------------------------------------
    def apply_mutation(self, name: str, **choices) -> Optional[dict]:
        """Apply the named mutation with explicit (previously sampled) choices."""
        method = getattr(self, name, None)
        if method is None:
            return None
        sig = inspect.signature(method)
        accepted = {
            k: v
            for k, v in choices.items()
            if k in sig.parameters
            or any(p.kind is inspect.Parameter.VAR_KEYWORD for p in sig.parameters.values())
        }
        return method(**accepted)

    def clone(self) -> "EvolvableModule":
        clone = type(self)(**self.init_dict)
        clone.load_state_dict(self.state_dict())
        try:
            clone = clone.to(self.device)
        except Exception:
            pass
        return clone

    # Subclasses with an internal rebuild implement this.
    def recreate_network(self) -> None:  # pragma: no cover - interface
        raise NotImplementedError

    def disable_mutations(self) -> None:
        self._mutations_disabled = True

    @property
    def mutations_enabled(self) -> bool:
        return not getattr(self, "_mutations_disabled", False)


class EvolvableWrapper(EvolvableModule):
    """Adapts a plain ``nn.Module`` into the evolvable interface.

    Architecture mutations are no-ops; cloning deep-copies the wrapped
    module. (Reference parity: ``agilerl/modules/base.py:760`` and
    ``modules/dummy.py:26`` ``DummyEvolvable``.)
    """

    def __init__(self, module: nn.Module, device: str = "cpu"):
        super().__init__(device)
        self.module = module.to(device) if device else module

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def clone(self) -> "EvolvableWrapper":
        return EvolvableWrapper(copy.deepcopy(self.module), device=self.device)


# Backwards-friendly alias matching the reference name.
DummyEvolvable = EvolvableWrapper


class ModuleDict(EvolvableModule):
    """Dict of evolvable modules keyed by agent-id (multi-agent nets).

    Mutations applied to the dict broadcast to every value; ``last_mutation``
    aggregates the first member's choices so a group replay is deterministic.
    (Reference parity: ``agilerl/modules/base.py:825``.)
    """

    def __init__(self, modules: Optional[Dict[str, EvolvableModule]] = None, device: str = "cpu"):
        super().__init__(device)
        built = {}
        for key, val in (modules or {}).items():
            if isinstance(val, dict) and "shared_with" in val:
                # identity-sharing marker (grouped agents share one module)
                val = built[val["shared_with"]]
            elif isinstance(val, dict) and "module_cls" in val:
                # self-describing child spec from init_dict (checkpoint rebuild)
                cls = val["module_cls"]
                init = dict(val["init_dict"])
                if "device" in init:
                    init["device"] = device
                val = cls(**init)
            built[key] = val
        self._modules_dict = nn.ModuleDict(built)

    def __getitem__(self, key: str) -> EvolvableModule:
        return self._modules_dict[key]

    def __setitem__(self, key: str, module: EvolvableModule) -> None:
        self._modules_dict[key] = module

    def __contains__(self, key: str) -> bool:
        return key in self._modules_dict

    def __iter__(self):
        return iter(self._modules_dict)

    def __len__(self) -> int:
        return len(self._modules_dict)

    def keys(self):
        return self._modules_dict.keys()

    def values(self):
        return self._modules_dict.values()

    def items(self):
        return self._modules_dict.items()

    def forward(self, key: str, *args, **kwargs):
        return self._modules_dict[key](*args, **kwargs)

    @property
    def mutation_methods(self) -> List[str]:
        first = next(iter(self._modules_dict.values()), None)
        return first.mutation_methods if first is not None else []

    def get_mutation_methods(self) -> Dict[str, MutationType]:
        # delegate to the first member (all members expose the same surface);
        # without this override the base-class type() lookup fails and
        # hpo.Mutations silently skips architecture mutations for MA agents
        first = next(iter(self._modules_dict.values()), None)
        return first.get_mutation_methods() if first is not None else {}

    def apply_mutation(self, name: str, **choices) -> Optional[dict]:
        out: Optional[dict] = None
        seen = set()  # grouped agents may share one module object
        for mod in self._modules_dict.values():
            if id(mod) in seen:
                continue
            seen.add(id(mod))
            if out is None:
                out = mod.apply_mutation(name, **choices)
                if isinstance(out, dict):
                    choices = {**choices, **out}
            else:
                mod.apply_mutation(name, **choices)
        self._last_mutation = (name, out or {})
        return out

    def clone(self) -> "ModuleDict":
        # preserve object sharing: keys mapping to the same module keep
        # mapping to ONE clone (grouped-agent net sharing)
        clones: Dict[int, EvolvableModule] = {}
        out = {}
        for k, v in self._modules_dict.items():
            if id(v) not in clones:
                clones[id(v)] = v.clone()
            out[k] = clones[id(v)]
        return ModuleDict(out, device=self.device)

    @property
    def init_dict(self) -> Dict[str, Any]:
        specs: Dict[str, Any] = {}
        seen: Dict[int, str] = {}
        for k, v in self._modules_dict.items():
            if id(v) in seen:
                specs[k] = {"shared_with": seen[id(v)]}
            else:
                seen[id(v)] = k
                specs[k] = {"module_cls": type(v), "init_dict": v.init_dict}
        return {"modules": specs, "device": self.device}


# ---------------------------------------------------------------------------
# Parameter preservation + module checkpointing
# ---------------------------------------------------------------------------

@torch.no_grad()
def preserve_parameters(old: nn.Module, new: nn.Module) -> nn.Module:
    """Copy overlapping parameter slices from ``old`` into ``new``.

    After an architecture mutation rebuilds a network, trained weights are
    retained wherever shapes overlap (reference:
    ``agilerl/networks/base.py:49`` / ``modules/mlp.py:317``).
    """
    old_params = dict(old.named_parameters())
    old_buffers = dict(old.named_buffers())
    for name, p_new in new.named_parameters():
        p_old = old_params.get(name)
        if p_old is None or p_old.dim() != p_new.dim():
            continue
        if p_old.shape == p_new.shape:
            p_new.copy_(p_old)
        else:
            slices = tuple(slice(0, min(a, b)) for a, b in zip(p_new.shape, p_old.shape))
            p_new[slices] = p_old[slices]
    for name, b_new in new.named_buffers():
        b_old = old_buffers.get(name)
        if b_old is None or b_old.shape != b_new.shape or b_old.dtype != b_new.dtype:
            continue
        b_new.copy_(b_old)
    return new


def module_checkpoint_dict(module: EvolvableModule) -> Dict[str, Any]:
    """Serializable description of an evolvable module (class + init + state)."""
    return {
        "module_cls": type(module),
        "init_dict": module.init_dict,
        "state_dict": {k: v.cpu() for k, v in module.state_dict().items()},
    }


def load_module_from_checkpoint(ckpt: Dict[str, Any], device: str = "cpu") -> EvolvableModule:
    cls = ckpt["module_cls"]
    init = dict(ckpt["init_dict"])
    if "device" in inspect.signature(cls.__init__).parameters:
        init["device"] = device
    module = cls(**init)
    module.load_state_dict(ckpt["state_dict"])
    return module.to(device)
