"""Mutation engine: none / architecture / parameter-noise / activation / RL-HP.

Reference parity: ``agilerl/hpo/mutation.py:207`` (Mutations — none :478,
architecture :488, RL-hyperparameter :527, activation :571, parameter
noise :634/:859).  New design: architecture mutations go through the
agent's :meth:`apply_architecture_mutation`, which samples the random
choices once on the policy network and replays them on every registered
network (targets/critics), then rebuilds optimizers — replacing the
reference's ``reinit_shared_networks`` / ``_reinit_from_mutated`` dance.
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch

from ..modules.base import MutationType

__all__ = ["Mutations"]


class Mutations:
    def __init__(
        self,
        no_mutation: float = 0.2,
        architecture: float = 0.2,
        new_layer_prob: float = 0.2,
        parameters: float = 0.2,
        activation: float = 0.0,
        rl_hp: float = 0.2,
        mutation_sd: float = 0.1,
        activation_selection: Optional[List[str]] = None,
        mutate_elite: bool = True,
        rand_seed: Optional[int] = None,
        device: str = "cpu",
        accelerator=None,
    ):
        if accelerator is not None:
            import warnings

            warnings.warn(
                "Mutations ignores `accelerator` (reference Accelerate-era "
                "kwarg): distributed mutation consensus runs over RCCL via "
                "agilerl_amd.parallel.", RuntimeWarning,
            )
        self.no_mutation = no_mutation
        self.architecture = architecture
        self.new_layer_prob = new_layer_prob
        self.parameters = parameters
        self.activation = activation
        self.rl_hp = rl_hp
        self.mutation_sd = mutation_sd
        self.activation_selection = activation_selection or ["ReLU", "ELU", "GELU"]
        self.mutate_elite = mutate_elite
        self.rng = np.random.default_rng(rand_seed)
        self.device = device

    # ------------------------------------------------------------------
    def mutation(self, population: List, pre_training: bool = False) -> List:
        """Apply one sampled mutation per agent (in place); returns population."""
        options = np.array(
            [self.no_mutation, self.architecture, self.parameters, self.activation, self.rl_hp]
        )
        if options.sum() <= 0:
            for agent in population:
                agent.mut = "None"
            return population
        probs = options / options.sum()
        for i, agent in enumerate(population):
            if i == 0 and not self.mutate_elite:
                agent.mut = "None"
                continue
            choice = self.rng.choice(5, p=probs)
            try:
                if choice == 0:
                    agent.mut = "None"
                elif choice == 1:
                    self.architecture_mutate(agent)
                elif choice == 2:
                    self.parameter_mutation(agent)
                elif choice == 3:
                    self.activation_mutation(agent)
                else:
                    self.rl_hyperparam_mutation(agent)
            except Exception as e:  # mutation must never kill training
                import warnings

                warnings.warn(
                    f"mutation failed on agent {getattr(agent, 'index', '?')}: "
                    f"{type(e).__name__}: {e}", RuntimeWarning,
                )
                agent.mut = f"Failed({type(e).__name__})"
        return population

    # ------------------------------------------------------------------
    def architecture_mutate(self, agent) -> None:
        methods = agent.mutation_methods
        if not methods:
            agent.mut = "None"
            return
        types = agent.policy_network.get_mutation_methods()
        layer_methods = [m for m in methods if types.get(m) == MutationType.LAYER]
        node_methods = [m for m in methods if types.get(m) == MutationType.NODE]
        if layer_methods and self.rng.random() < self.new_layer_prob:
            method = layer_methods[int(self.rng.integers(len(layer_methods)))]
        elif node_methods:
            method = node_methods[int(self.rng.integers(len(node_methods)))]
        else:
            method = methods[int(self.rng.integers(len(methods)))]
        agent.apply_architecture_mutation(method)
        agent.mut = method

    @torch.no_grad()
    def parameter_mutation(self, agent) -> None:
        """Gaussian noise on 10% of the policy network's weights."""
        policy = agent.policy_network
        for param in policy.parameters():
            if param.dim() < 1:
                continue
            mask = torch.rand_like(param) < 0.1
            noise = torch.randn_like(param) * self.mutation_sd
            param.add_(noise * mask)
        # mirror into shared networks (e.g. targets keep polyak pace naturally)
        agent.mutation_hook()
        agent.mut = "param"

    def activation_mutation(self, agent) -> None:
        activation = str(self.rng.choice(self.activation_selection))
        methods = [m for m in agent.mutation_methods if m.endswith("change_activation")]
        if not methods:
            agent.mut = "None"
            return
        for m in methods:
            agent.apply_architecture_mutation(m, activation=activation)
        agent.mut = "act"

    def rl_hyperparam_mutation(self, agent) -> None:
        name = agent.hp_config.sample(self.rng)
        if name is None:
            agent.mut = "None"
            return
        spec = agent.hp_config[name]
        old = getattr(agent, name)
        new = spec.mutate(old, self.rng)
        setattr(agent, name, new)
        # learning-rate changes propagate into live optimizers
        for cfg in agent.registry.optimizer_configs:
            if cfg.lr_name == name:
                getattr(agent, cfg.name).update_lr(float(new))
        agent.mut = name
