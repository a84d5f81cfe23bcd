"""Tournament + mutation engine tests."""

import numpy as np
import torch

from agilerl_amd.algorithms.dqn import DQN
from agilerl_amd.hpo import Mutations, TournamentSelection
from agilerl_amd.spaces import Box, Discrete


def make_pop(n=4):
    pop = DQN.population(n, Box(-1, 1, (4,)), Discrete(2))
    for i, agent in enumerate(pop):
        agent.fitness.append(float(i))
    return pop


class TestTournament:
    def test_elite_kept(self):
        pop = make_pop(4)
        tour = TournamentSelection(tournament_size=2, elitism=True)
        elite, new_pop = tour.select(pop)
        assert elite is pop[3]
        assert len(new_pop) == 4
        x = torch.randn(2, 4)
        assert torch.allclose(new_pop[0].actor(x), pop[3].actor(x))

    def test_plan_indices_valid(self):
        tour = TournamentSelection(tournament_size=3, elitism=True)
        fits = np.array([0.0, 5.0, 3.0, 1.0])
        plan = tour.compute_plan(fits, pop_size=6)
        assert plan[0] == 1
        assert all(0 <= p < 4 for p in plan)
        assert len(plan) == 6

    def test_indices_reassigned(self):
        pop = make_pop(3)
        _, new_pop = TournamentSelection(2, True).select(pop)
        assert [a.index for a in new_pop] == [0, 1, 2]


class TestMutations:
    def test_architecture_mutation_applies(self):
        pop = make_pop(3)
        muts = Mutations(no_mutation=0, architecture=1.0, parameters=0, activation=0, rl_hp=0, rand_seed=0)
        new_pop = muts.mutation(pop)
        assert all(a.mut not in ("None",) for a in new_pop)
        x = torch.randn(2, 4)
        for a in new_pop:
            assert a.actor(x).shape == (2, 2)
            # target mirrored
            assert torch.allclose(a.actor(x), a.actor_target(x))

    def test_rl_hp_mutation(self):
        pop = make_pop(2)
        muts = Mutations(no_mutation=0, architecture=0, parameters=0, activation=0, rl_hp=1.0, rand_seed=0)
        old = {id(a): (a.lr, a.batch_size, a.learn_step) for a in pop}
        new_pop = muts.mutation(pop)
        for a in new_pop:
            assert a.mut in ("lr", "batch_size", "learn_step")
            assert (a.lr, a.batch_size, a.learn_step) != old[id(a)]
            if a.mut == "lr":
                assert a.optimizer.param_groups[0]["lr"] == a.lr

    def test_parameter_mutation_changes_weights(self):
        pop = make_pop(1)
        w_before = pop[0].actor.head_net.model[0].weight.detach().clone()
        muts = Mutations(no_mutation=0, architecture=0, parameters=1.0, activation=0, rl_hp=0,
                         mutation_sd=0.5, rand_seed=0)
        muts.mutation(pop)
        w_after = pop[0].actor.head_net.model[0].weight.detach()
        assert not torch.equal(w_before, w_after)

    def test_no_mutation(self):
        pop = make_pop(2)
        muts = Mutations(no_mutation=1.0, architecture=0, parameters=0, activation=0, rl_hp=0)
        new_pop = muts.mutation(pop)
        assert all(a.mut == "None" for a in new_pop)

    def test_mutated_clone_learns(self):
        """A mutated clone must still be trainable (optimizer points at live params)."""
        pop = make_pop(2)
        muts = Mutations(no_mutation=0, architecture=1.0, parameters=0, activation=0, rl_hp=0, rand_seed=3)
        tour = TournamentSelection(2, True)
        _, new_pop = tour.select(pop)
        new_pop = muts.mutation(new_pop)
        agent = new_pop[1]
        batch = {
            "obs": torch.randn(32, 4),
            "action": torch.randint(0, 2, (32,)),
            "reward": torch.randn(32),
            "next_obs": torch.randn(32, 4),
            "done": torch.zeros(32),
        }
        w_before = [p.detach().clone() for p in agent.actor.parameters()]
        agent.learn(batch)
        changed = any(
            not torch.equal(b, p.detach()) for b, p in zip(w_before, agent.actor.parameters())
        )
        assert changed
