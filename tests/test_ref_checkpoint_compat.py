"""Reference-checkpoint cross-compatibility.

Builds a checkpoint file in the REFERENCE's exact on-disk layout
(``agilerl/algorithms/core/base.py:315-372``: flat attribute dict +
``network_info`` with ``{name}_cls/{name}_init_dict/{name}_state_dict``
keys) whose pickled class references point at ``agilerl.*`` and
``gymnasium.spaces.*`` module paths — exactly what a reference-written
file contains — and asserts our loader maps them onto this package's
classes and restores weights bit-for-bit.
"""

import sys
import types

import dill
import numpy as np
import pytest
import torch


def _fake_module(path):
    """Register an empty module chain under ``path`` in sys.modules."""
    created = []
    parts = path.split(".")
    for i in range(1, len(parts) + 1):
        name = ".".join(parts[:i])
        if name not in sys.modules:
            mod = types.ModuleType(name)
            sys.modules[name] = mod
            created.append(name)
    return created


class _RefPickleCtx:
    """Temporarily alias our classes under reference module paths so dill
    writes `agilerl.networks.q_networks.QNetwork` etc. into the stream,
    then removes the fakes so loading MUST go through the compat mapping."""

    ALIASES = [
        ("agilerl.networks.q_networks", "QNetwork",
         "agilerl_amd.networks.q_networks"),
        ("agilerl.modules.mlp", "EvolvableMLP", "agilerl_amd.modules.mlp"),
        ("gymnasium.spaces.box", "Box", "agilerl_amd.spaces"),
        ("gymnasium.spaces.discrete", "Discrete", "agilerl_amd.spaces"),
    ]

    def __enter__(self):
        import importlib

        self._created = []
        self._patched = []
        for fake_path, cls_name, real_path in self.ALIASES:
            self._created += _fake_module(fake_path)
            real_cls = getattr(importlib.import_module(real_path), cls_name)
            fake = sys.modules[fake_path]
            setattr(fake, cls_name, real_cls)
            # point the class's own metadata at the fake path so pickle
            # records it (save originals for restore)
            self._patched.append((real_cls, real_cls.__module__))
            real_cls.__module__ = fake_path
        return self

    def __exit__(self, *exc):
        for cls, orig in self._patched:
            cls.__module__ = orig
        for name in self._created:
            sys.modules.pop(name, None)
        return False


@pytest.fixture()
def reference_ckpt_file(tmp_path):
    """A DQN checkpoint written in the reference layout under reference
    module paths."""
    from agilerl_amd.algorithms.dqn import DQN
    from agilerl_amd.spaces import Box, Discrete

    torch.manual_seed(5)
    agent = DQN(Box(-1, 1, (4,)), Discrete(2), lr=3e-4, batch_size=32, gamma=0.97)
    agent.fitness = [1.0, 7.5]
    agent.steps = [123]

    # reference layout: flat attributes + network_info flat-key modules
    ckpt = {
        "agilerl_version": "2.0.0",
        "algo": "DQN",
        "lr": agent.lr,
        "batch_size": agent.batch_size,
        "gamma": agent.gamma,
        "tau": agent.tau,
        "learn_step": agent.learn_step,
        "fitness": list(agent.fitness),
        "steps": list(agent.steps),
        "observation_space": agent.observation_space,
        "action_space": agent.action_space,
        "mut": "None",
        "index": 0,
        "double": False,
        "network_info": {
            "modules": {
                "actor_cls": type(agent.actor),
                "actor_init_dict": dict(agent.actor.init_dict,
                                        reference_only_field="dropme"),
                "actor_state_dict": {k: v.cpu() for k, v in agent.actor.state_dict().items()},
                "actor_module_dict_cls": None,
                "actor_target_cls": type(agent.actor_target),
                "actor_target_init_dict": dict(agent.actor_target.init_dict),
                "actor_target_state_dict": {
                    k: v.cpu() for k, v in agent.actor_target.state_dict().items()
                },
                "actor_target_module_dict_cls": None,
            },
            "optimizers": {
                "optimizer_state_dict": agent.optimizer.state_dict(),
            },
            "network_names": ["actor", "actor_target"],
            "optimizer_names": ["optimizer"],
        },
    }
    path = tmp_path / "reference_dqn.pt"
    # plain pickle records classes BY REFERENCE (module path + name), the
    # stream shape our compat mapping targets (dill with byref, and any
    # pickle-protocol writer, produce the same GLOBAL opcodes)
    import pickle

    with _RefPickleCtx():
        torch.save(ckpt, str(path), pickle_module=pickle)
    # the network classes must be recorded under the REFERENCE paths, so
    # loading can only succeed through the compat class mapping
    raw = path.read_bytes()
    assert b"agilerl.networks.q_networks" in raw
    assert b"agilerl_amd.networks" not in raw
    return str(path), agent


class TestReferenceCheckpointCompat:
    def test_load_checkpoint_into_existing_agent(self, reference_ckpt_file):
        from agilerl_amd.algorithms.dqn import DQN
        from agilerl_amd.spaces import Box, Discrete

        path, src = reference_ckpt_file
        torch.manual_seed(99)  # different init
        dst = DQN(Box(-1, 1, (4,)), Discrete(2))
        dst.load_checkpoint(path)
        x = torch.randn(6, 4, generator=torch.Generator().manual_seed(1))
        torch.testing.assert_close(dst.actor(x), src.actor(x))
        torch.testing.assert_close(dst.actor_target(x), src.actor_target(x))
        assert dst.lr == src.lr
        assert dst.batch_size == src.batch_size
        assert dst.gamma == src.gamma
        assert dst.fitness == [1.0, 7.5]
        assert dst.steps == [123]

    def test_classmethod_load_rebuilds_agent(self, reference_ckpt_file):
        from agilerl_amd.algorithms.core.base import EvolvableAlgorithm
        from agilerl_amd.algorithms.dqn import DQN

        path, src = reference_ckpt_file
        agent = EvolvableAlgorithm.load(path)
        assert isinstance(agent, DQN)
        x = torch.randn(6, 4, generator=torch.Generator().manual_seed(2))
        torch.testing.assert_close(agent.actor(x), src.actor(x))
        assert agent.gamma == src.gamma
        # loaded agent must remain fully functional: learn + clone + save
        batch = {
            "obs": torch.randn(8, 4), "action": torch.randint(0, 2, (8, 1)),
            "reward": torch.randn(8, 1), "next_obs": torch.randn(8, 4),
            "done": torch.zeros(8, 1),
        }
        agent.learn(batch)
        clone = agent.clone(3)
        assert clone.index == 3

    def test_unknown_reference_class_raises_informatively(self):
        from agilerl_amd.utils.ref_compat import _resolve

        with pytest.raises(ModuleNotFoundError, match="ExoticNet"):
            _resolve("ExoticNet", "agilerl.networks.exotic")

    def test_gymnasium_space_state_maps_onto_ours(self, tmp_path):
        """A pickled gymnasium Box restores its state onto our Box through
        find_class mapping; gym attribute names (_shape) are normalized."""
        import io

        from agilerl_amd.spaces import Box as OurBox
        from agilerl_amd.utils.ref_compat import (
            _gym_space_to_ours,
            load_checkpoint_file,
        )

        created = _fake_module("gymnasium.spaces.box")

        class Box:  # stand-in with gymnasium's pickled attribute layout
            pass

        Box.__module__ = "gymnasium.spaces.box"
        Box.__qualname__ = "Box"
        sys.modules["gymnasium.spaces.box"].Box = Box
        gym_box = Box()
        gym_box.__dict__.update(
            low=np.full(3, -1.0, np.float32),
            high=np.full(3, 1.0, np.float32),
            _shape=(3,), dtype=np.float32,
            bounded_below=np.ones(3, bool), bounded_above=np.ones(3, bool),
        )
        path = tmp_path / "space.pt"
        import pickle

        torch.save({"space": gym_box}, str(path), pickle_module=pickle)
        for name in created:
            sys.modules.pop(name, None)
        out = load_checkpoint_file(str(path))["space"]
        assert isinstance(out, OurBox)
        out = _gym_space_to_ours(out)
        assert out.shape == (3,)
        np.testing.assert_array_equal(out.low, np.full(3, -1.0, np.float32))

    def test_round_trip_own_format_still_works(self, tmp_path):
        from agilerl_amd.algorithms.dqn import DQN
        from agilerl_amd.spaces import Box, Discrete

        torch.manual_seed(0)
        agent = DQN(Box(-1, 1, (4,)), Discrete(2))
        p = tmp_path / "own.pt"
        agent.save_checkpoint(str(p))
        torch.manual_seed(1)
        other = DQN(Box(-1, 1, (4,)), Discrete(2))
        other.load_checkpoint(str(p))
        x = torch.randn(3, 4)
        torch.testing.assert_close(other.actor(x), agent.actor(x))
