"""hipGraph x mutation correctness sweep.

An incorrect graph adoption after evolution silently corrupts training:
the captured graph is bound to the OLD parameter tensors, so a stale
graph replays updates into dead memory.  This sweep pins, on a real GPU:

- graphed learn == eager learn (full-batch, single epoch => the update is
  permutation-invariant and must match an eager twin parameter-for-
  parameter);
- architecture mutation -> graph cleared -> recapture -> still matches an
  identically-mutated eager twin;
- lr mutation -> graph rebuilt with the new lr (the capturable Adam bakes
  lr at capture time);
- the adopt_agent_state fast path (bench.py's evolution round) keeps a
  recaptured graph valid.
"""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _flat_rollout(n, obs_dim, n_actions, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    return {
        "obs": torch.randn(n, obs_dim, generator=g).to(DEV),
        "action": torch.randint(0, n_actions, (n,), generator=g).to(DEV),
        "log_prob": torch.randn(n, generator=g).mul(0.1).to(DEV),
        "advantages": torch.randn(n, generator=g).to(DEV),
        "returns": torch.randn(n, generator=g).to(DEV),
        "value": torch.randn(n, generator=g).to(DEV),
        "done": torch.zeros(n).to(DEV),
        "reward": torch.zeros(n).to(DEV),
    }


def _make_ppo_pair(n):
    """(graphed agent, eager twin) with identical weights; full-batch
    single-epoch config so the two paths compute the same update."""
    from agilerl_amd.algorithms.ppo import PPO
    from agilerl_amd.spaces import Box, Discrete

    agents = []
    for _ in range(2):
        torch.manual_seed(7)
        agents.append(PPO(
            Box(-1, 1, (8,)), Discrete(4), batch_size=n, update_epochs=1,
            device=DEV, net_config={"arch": "mlp", "hidden_size": [32, 32]},
        ))
    graphed, eager = agents
    eager.target_kl = 1e9  # forces the non-graphed learn branch
    return graphed, eager


def _assert_params_close(a, b, rtol=2e-3, atol=2e-4):
    for name in ("actor", "critic"):
        pa = dict(getattr(a, name).named_parameters())
        pb = dict(getattr(b, name).named_parameters())
        assert pa.keys() == pb.keys()
        for k in pa:
            torch.testing.assert_close(pa[k], pb[k], rtol=rtol, atol=atol,
                                       msg=lambda m: f"{name}.{k}: {m}")


class TestPPOGraphMutationSweep:
    N = 256

    def test_graphed_matches_eager_then_arch_mutation_recapture(self):
        graphed, eager = _make_ppo_pair(self.N)
        flat = _flat_rollout(self.N, 8, 4, seed=1)
        graphed.learn(dict(flat))
        assert graphed._learn_graph is not None, "graph must have captured"
        eager.learn(dict(flat))
        _assert_params_close(graphed, eager)

        # identical architecture mutation on both: pin BOTH the sampled
        # layer choice (add_node draws it from a non-torch RNG) and the
        # torch seed (new-node weight init)
        for agent in (graphed, eager):
            torch.manual_seed(123)
            torch.cuda.manual_seed_all(123)
            agent.apply_architecture_mutation(
                "encoder.add_node", hidden_layer=0, numb_new_nodes=16)
        assert graphed._learn_graph is None, "mutation hook must clear the graph"
        flat2 = _flat_rollout(self.N, 8, 4, seed=2)
        graphed.learn(dict(flat2))   # recapture with the mutated architecture
        assert graphed._learn_graph is not None
        eager.learn(dict(flat2))
        _assert_params_close(graphed, eager)

        # and replays keep working (second learn on the captured graph)
        flat3 = _flat_rollout(self.N, 8, 4, seed=3)
        graphed.learn(dict(flat3))
        eager.learn(dict(flat3))
        _assert_params_close(graphed, eager)

    def test_lr_mutation_rebuilds_graph_with_new_lr(self):
        graphed, eager = _make_ppo_pair(self.N)
        flat = _flat_rollout(self.N, 8, 4, seed=4)
        graphed.learn(dict(flat))
        eager.learn(dict(flat))
        _assert_params_close(graphed, eager)

        # lr mutation: the graphed capturable-Adam baked the old lr, so the
        # graph must be invalidated (the adopt_agent_state bail path)
        for agent in (graphed, eager):
            agent.lr = 1e-2
            agent._reinit_optimizers()
        graphed._clear_learn_graph()
        flat2 = _flat_rollout(self.N, 8, 4, seed=5)
        graphed.learn(dict(flat2))
        eager.learn(dict(flat2))
        _assert_params_close(graphed, eager, rtol=5e-3, atol=5e-4)
        # the big lr must actually have moved the weights substantially
        before = dict(_make_ppo_pair(self.N)[0].actor.named_parameters())
        after = dict(graphed.actor.named_parameters())
        deltas = [(after[k] - before[k]).abs().max().item() for k in after]
        assert max(deltas) > 1e-3, "lr mutation had no effect on the update"

    def test_adopt_agent_state_keeps_recaptured_graph_valid(self):
        """bench.py's evolution round: offspring adopted into the existing
        object; the captured graph must keep producing the same updates as
        an eager twin that went through the same adoption."""
        from agilerl_amd.parallel.population_runtime import adopt_agent_state

        graphed, eager = _make_ppo_pair(self.N)
        flat = _flat_rollout(self.N, 8, 4, seed=6)
        graphed.learn(dict(flat))
        eager.learn(dict(flat))

        # same-arch offspring with different weights (both sides identical)
        torch.manual_seed(99)
        child_g = graphed.clone(1)
        with torch.no_grad():
            for p in child_g.actor.parameters():
                p.add_(0.01)
        torch.manual_seed(99)
        child_e = eager.clone(1)
        with torch.no_grad():
            for p in child_e.actor.parameters():
                p.add_(0.01)
        assert adopt_agent_state(graphed, child_g)
        assert adopt_agent_state(eager, child_e)
        assert graphed._learn_graph is not None, "same-arch adoption keeps the graph"

        flat2 = _flat_rollout(self.N, 8, 4, seed=7)
        graphed.learn(dict(flat2))
        eager.learn(dict(flat2))
        _assert_params_close(graphed, eager)


class TestDQNGraphMutationSweep:
    def _pair(self):
        from agilerl_amd.algorithms.dqn import DQN
        from agilerl_amd.spaces import Box, Discrete

        agents = []
        for graphs in (True, False):
            torch.manual_seed(11)
            agents.append(DQN(
                Box(-1, 1, (6,)), Discrete(3), batch_size=128,
                cudagraphs=graphs, device=DEV,
                net_config={"arch": "mlp", "hidden_size": [32]},
            ))
        return agents

    def _batch(self, seed):
        g = torch.Generator().manual_seed(seed)
        return {
            "obs": torch.randn(128, 6, generator=g).to(DEV),
            "action": torch.randint(0, 3, (128, 1), generator=g).to(DEV),
            "reward": torch.randn(128, 1, generator=g).to(DEV),
            "next_obs": torch.randn(128, 6, generator=g).to(DEV),
            "done": torch.zeros(128, 1).to(DEV),
        }

    def test_graphed_update_survives_arch_mutation(self):
        graphed, eager = self._pair()
        for seed in (1, 2):
            graphed.learn(self._batch(seed))
            eager.learn(self._batch(seed))
        for (ka, pa), (kb, pb) in zip(
            graphed.actor.named_parameters(), eager.actor.named_parameters()
        ):
            torch.testing.assert_close(pa, pb, rtol=2e-3, atol=2e-4)

        for agent in (graphed, eager):
            torch.manual_seed(321)
            torch.cuda.manual_seed_all(321)
            agent.apply_architecture_mutation("encoder.add_layer")
        assert [tuple(p.shape) for p in graphed.actor.parameters()] == \
            [tuple(p.shape) for p in eager.actor.parameters()]
        for seed in (3, 4):
            graphed.learn(self._batch(seed))
            eager.learn(self._batch(seed))
        for (ka, pa), (kb, pb) in zip(
            graphed.actor.named_parameters(), eager.actor.named_parameters()
        ):
            torch.testing.assert_close(pa, pb, rtol=5e-3, atol=5e-4)
