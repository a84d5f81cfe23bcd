"""Mutation-invariant sweep: every algorithm x every mutation method.

For each registered classic-RL algorithm: construct an agent, apply each
architecture mutation, and verify (a) all group networks stay
structurally consistent, (b) a learn step still runs, (c) the clone of
the mutated agent is functionally identical.  This is the broad
regression net over the evolvable object model (SURVEY hard-part #2).
"""

import numpy as np
import pytest
import torch

from agilerl_amd.algorithms import CQN, DDPG, DQN, PPO, TD3, RainbowDQN
from agilerl_amd.spaces import Box, Discrete

OBS = Box(-1.0, 1.0, (6,))
DISC = Discrete(3)
CONT = Box(-1.0, 1.0, (2,))

NET = {"arch": "mlp", "hidden_size": [16, 16]}


def make_batch(agent, continuous: bool):
    B = 16
    action = torch.rand(B, 2) * 2 - 1 if continuous else torch.randint(0, 3, (B,))
    return {
        "obs": torch.randn(B, 6),
        "action": action,
        "reward": torch.randn(B),
        "next_obs": torch.randn(B, 6),
        "done": torch.zeros(B),
    }


def check_learn(agent, continuous: bool):
    loss = agent.learn(make_batch(agent, continuous))
    assert np.isfinite(loss)


CASES = [
    (DQN, DISC, False),
    (CQN, DISC, False),
    (RainbowDQN, DISC, False),
    (DDPG, CONT, True),
    (TD3, CONT, True),
]


@pytest.mark.parametrize("cls,act_space,continuous", CASES,
                         ids=[c[0].__name__ for c in CASES])
def test_every_mutation_preserves_learnability(cls, act_space, continuous):
    base = cls(OBS, act_space, net_config=dict(NET), batch_size=16)
    for method in base.mutation_methods:
        agent = cls(OBS, act_space, net_config=dict(NET), batch_size=16)
        agent.apply_architecture_mutation(method)
        # group consistency: every shared (target) net matches its eval net
        for group in agent.registry.groups:
            ev = getattr(agent, group.eval_network)
            n_ev = sum(p.numel() for p in ev.parameters())
            for shared in group.shared_networks:
                sh = getattr(agent, shared)
                n_sh = sum(p.numel() for p in sh.parameters())
                assert n_ev == n_sh, f"{cls.__name__}.{method}: {group.eval_network} vs {shared}"
        check_learn(agent, continuous)
        # clone of the mutated agent behaves identically
        clone = agent.clone(index=9)
        x = torch.randn(4, 6)
        pol, cpol = agent.policy_network, clone.policy_network
        pol.eval(), cpol.eval()
        torch.testing.assert_close(pol(x), cpol(x))


@pytest.mark.parametrize("method", PPO(OBS, DISC, net_config=dict(NET)).mutation_methods)
def test_ppo_mutations(method):
    from agilerl_amd.components import RolloutBuffer
    from agilerl_amd.envs import CartPoleVecEnv
    from agilerl_amd.rollouts.on_policy import collect_rollouts

    env = CartPoleVecEnv(4, seed=0)
    agent = PPO(env.observation_space, env.action_space, net_config=dict(NET),
                learn_step=8, batch_size=16)
    agent.apply_architecture_mutation(method)
    buf = RolloutBuffer(8, 4, gamma=agent.gamma, gae_lambda=agent.gae_lambda)
    collect_rollouts(agent, env, buf, 8)
    stats = agent.learn(buf)
    assert np.isfinite(stats["policy_loss"])


# ---------------------------------------------------------------------------
# multi-agent sweep: ModuleDict groups + shared-agent dedup under mutation
# ---------------------------------------------------------------------------

def _ma_agent(cls, shared=False):
    from agilerl_amd.envs.mpe import SpeakerListenerVecEnv

    env = SpeakerListenerVecEnv(num_envs=2, seed=0)
    kw = {}
    if shared:
        kw["shared_agent_groups"] = [["speaker_0"], ["listener_0"]]
    agent = cls(
        env.observation_spaces, env.action_spaces, agent_ids=env.agents,
        batch_size=16, net_config={"arch": "mlp", "hidden_size": [16]}, **kw,
    )
    return env, agent


def _ma_offpolicy_batch(env, agent):
    from agilerl_amd.components import ReplayBuffer

    buf = ReplayBuffer(200)
    obs, _ = env.reset()
    for _ in range(24):
        env_actions, raw = agent.get_action(obs)
        next_obs, rewards, term, trunc, _ = env.step(env_actions)
        buf.add(obs=obs, action=raw,
                reward={a: rewards[a] for a in env.agents},
                next_obs=next_obs,
                done={a: term[a].astype(np.float32) for a in env.agents})
        obs = next_obs
    return buf.sample(16)


MA_OFFPOLICY = []
try:
    from agilerl_amd.algorithms import MADDPG, MATD3

    MA_OFFPOLICY = [MADDPG, MATD3]
except ImportError:  # pragma: no cover
    pass


@pytest.mark.parametrize("cls", MA_OFFPOLICY, ids=lambda c: c.__name__)
def test_multiagent_mutation_sweep(cls):
    env, base = _ma_agent(cls)
    for method in base.mutation_methods:
        env, agent = _ma_agent(cls)
        agent.apply_architecture_mutation(method)
        # every target ModuleDict mirrors its eval ModuleDict per agent id
        for group in agent.registry.groups:
            ev = getattr(agent, group.eval_network)
            for shared in group.shared_networks:
                sh = getattr(agent, shared)
                for aid in ev.keys():
                    n_ev = sum(p.numel() for p in ev[aid].parameters())
                    n_sh = sum(p.numel() for p in sh[aid].parameters())
                    assert n_ev == n_sh, f"{cls.__name__}.{method}[{aid}]"
        loss = agent.learn(_ma_offpolicy_batch(env, agent))
        assert np.isfinite(loss)
        clone = agent.clone(index=3)
        obs, _ = env.reset()
        a1, _ = agent.get_action(obs, training=False)
        a2, _ = clone.get_action(obs, training=False)
        for aid in env.agents:
            np.testing.assert_array_equal(a1[aid], a2[aid])


def test_ippo_mutation_sweep():
    from agilerl_amd.algorithms import IPPO
    from agilerl_amd.components import RolloutBuffer
    from agilerl_amd.training.train_multi_agent_on_policy import _collect_ma_rollout

    env, base = _ma_agent(IPPO)
    for method in base.mutation_methods:
        env, agent = _ma_agent(IPPO)
        agent.learn_step = 8
        agent.apply_architecture_mutation(method)
        bufs = {
            aid: RolloutBuffer(8, env.num_envs, gamma=agent.gamma,
                               gae_lambda=agent.gae_lambda)
            for aid in env.agents
        }
        _collect_ma_rollout(agent, env, bufs, 8, None)
        stats = agent.learn(bufs)
        assert np.isfinite(stats["policy_loss"])


def test_maddpg_shared_groups_stay_shared_after_mutation():
    from agilerl_amd.algorithms import IPPO

    env, agent = _ma_agent(IPPO, shared=False)
    # same-space agents sharing one module: build a 2-listener env stand-in
    obs_sp = {"a_0": env.observation_spaces["listener_0"],
              "a_1": env.observation_spaces["listener_0"]}
    act_sp = {"a_0": env.action_spaces["listener_0"],
              "a_1": env.action_spaces["listener_0"]}
    agent = IPPO(obs_sp, act_sp, agent_ids=["a_0", "a_1"],
                 shared_agent_groups=[["a_0", "a_1"]],
                 net_config={"arch": "mlp", "hidden_size": [16]})
    assert agent.actors["a_0"] is agent.actors["a_1"]
    for method in agent.mutation_methods:
        agent.apply_architecture_mutation(method)
        assert agent.actors["a_0"] is agent.actors["a_1"], method
        assert agent.critics["a_0"] is agent.critics["a_1"], method


@pytest.mark.parametrize("cls_name", ["NeuralUCB", "NeuralTS"])
def test_bandit_mutation_sweep(cls_name):
    """Neural bandits: every mutation keeps the confidence machinery and
    learn path functional (reference mutation.py:1196 grad reinit)."""
    import agilerl_amd.algorithms as algos

    cls = getattr(algos, cls_name)
    ctx_dim = 8
    base = cls(Box(-1.0, 1.0, (ctx_dim,)), Discrete(4),
               net_config={"arch": "mlp", "hidden_size": [16]})
    for method in base.mutation_methods:
        agent = cls(Box(-1.0, 1.0, (ctx_dim,)), Discrete(4),
                    net_config={"arch": "mlp", "hidden_size": [16]})
        agent.apply_architecture_mutation(method)
        context = np.random.randn(4, ctx_dim).astype(np.float32)
        arm = agent.get_action(context)
        assert 0 <= arm < 4
        batch = {"obs": torch.randn(16, ctx_dim), "reward": torch.rand(16)}
        loss = agent.learn(batch)
        assert np.isfinite(loss), method
        clone = agent.clone(index=3)
        x = torch.randn(2, ctx_dim)
        torch.testing.assert_close(agent.actor(x), clone.actor(x))


@pytest.mark.parametrize("cls_name", ["MADDPG", "MATD3", "IPPO"])
def test_hpo_mutations_actually_apply_to_multiagent(cls_name):
    """Regression: hpo.Mutations must reach multi-agent ModuleDicts — the
    type-level lookup in get_mutation_methods used to fail and silently
    skip every architecture mutation (surfaced as Failed(AttributeError))."""
    import agilerl_amd.algorithms as algos
    from agilerl_amd.envs.mpe import SpeakerListenerVecEnv
    from agilerl_amd.hpo import Mutations

    cls = getattr(algos, cls_name)
    torch.manual_seed(0), np.random.seed(0)
    env = SpeakerListenerVecEnv(num_envs=2, seed=0)
    pop = cls.population(4, env.observation_spaces, env.action_spaces,
                            agent_ids=env.agents,
                            net_config={"arch": "mlp", "hidden_size": [16]})
    muts = Mutations(no_mutation=0.0, architecture=1.0, parameters=0.0,
                     activation=0.0, rl_hp=0.0, rand_seed=0)
    pop = muts.mutation(pop)
    applied = [a.mut for a in pop]
    assert not any(m.startswith("Failed") for m in applied), applied
    assert any(m != "None" for m in applied), applied


class TestArchTypeMutationSweep:
    """Architecture mutations across every encoder family (the reference
    mutation suite exercises CNN/MultiInput/SimBa too, not just MLP)."""

    def _agent_for(self, arch):
        from agilerl_amd.algorithms.dqn import DQN
        from agilerl_amd.spaces import Box, DictSpace, Discrete

        torch.manual_seed(0)
        if arch == "cnn":
            space = Box(0, 255, (3, 16, 16))
            cfg = {"arch": "cnn", "channel_size": [8, 8], "kernel_size": [3, 3],
                   "stride_size": [1, 1]}
        elif arch == "simba":
            space = Box(-1, 1, (8,))
            cfg = {"arch": "simba", "hidden_size": 32, "num_blocks": 1}
        elif arch == "multi_input":
            space = DictSpace({"vec": Box(-1, 1, (6,)), "img": Box(0, 255, (1, 8, 8))})
            cfg = {"arch": "multi_input"}
        else:
            space = Box(-1, 1, (8,))
            cfg = {"arch": "mlp", "hidden_size": [16, 16]}
        return DQN(space, Discrete(3), net_config=cfg, batch_size=8), space

    def _batch(self, space):
        from agilerl_amd.spaces import DictSpace

        def sample(n):
            if isinstance(space, DictSpace):
                return {k: torch.as_tensor(
                    np.stack([space.spaces[k].sample() for _ in range(n)])
                ).float() for k in space.spaces}
            return torch.as_tensor(
                np.stack([space.sample() for _ in range(n)])).float()

        return {
            "obs": sample(8),
            "action": torch.randint(0, 3, (8, 1)),
            "reward": torch.randn(8, 1),
            "next_obs": sample(8),
            "done": torch.zeros(8, 1),
        }

    @pytest.mark.parametrize("arch", ["mlp", "cnn", "simba", "multi_input"])
    def test_every_architecture_mutation_keeps_agent_trainable(self, arch):
        agent, space = self._agent_for(arch)
        methods = [m for m in agent.mutation_methods if "activation" not in m]
        assert methods, f"{arch}: no architecture mutations exposed"
        applied = 0
        for method in methods:
            agent.apply_architecture_mutation(method)
            applied += 1
            loss = agent.learn(self._batch(space))
            assert np.isfinite(loss)
            # targets must track the mutated eval net structurally
            sd_a = agent.actor.state_dict()
            sd_t = agent.actor_target.state_dict()
            assert set(sd_a.keys()) == set(sd_t.keys())
            for k in sd_a:
                assert sd_a[k].shape == sd_t[k].shape, (arch, method, k)
        assert applied == len(methods)
        # clone + checkpoint survive the fully-mutated architecture
        clone = agent.clone(5)
        x = self._batch(space)["obs"]
        if isinstance(x, dict):
            torch.testing.assert_close(clone.actor(x), agent.actor(x))
        else:
            torch.testing.assert_close(clone.actor(x), agent.actor(x))

    def test_lstm_network_mutations(self):
        from agilerl_amd.modules.lstm import EvolvableLSTM

        torch.manual_seed(1)
        lstm = EvolvableLSTM(input_size=6, num_outputs=4, hidden_state_size=16)
        for method in list(lstm.mutation_methods):
            if "activation" in method:
                continue
            lstm.apply_mutation(method)
            out = lstm(torch.randn(3, 5, 6))
            assert out.shape[-1] == 4
            assert torch.isfinite(out).all()


class TestArchSweepOtherAlgos:
    """The replay-desync bug class hits any algorithm with target nets —
    pin Rainbow(cnn) and DDPG(multi_input) too."""

    def test_rainbow_cnn_mutations(self):
        from agilerl_amd.algorithms.dqn_rainbow import RainbowDQN
        from agilerl_amd.spaces import Box, Discrete

        torch.manual_seed(0)
        agent = RainbowDQN(
            Box(0, 255, (3, 16, 16)), Discrete(4), batch_size=8, n_step=1,
            net_config={"arch": "cnn", "channel_size": [8, 8],
                        "kernel_size": [3, 3], "stride_size": [1, 1]},
        )
        batch = {
            "obs": torch.rand(8, 3, 16, 16) * 255,
            "action": torch.randint(0, 4, (8, 1)),
            "reward": torch.randn(8, 1),
            "next_obs": torch.rand(8, 3, 16, 16) * 255,
            "done": torch.zeros(8, 1),
            "weights": torch.ones(8),
            "idxs": torch.arange(8),
        }
        for method in [m for m in agent.mutation_methods if "activation" not in m]:
            agent.apply_architecture_mutation(method)
            loss = agent.learn(batch)
            assert np.isfinite(float(loss if not isinstance(loss, tuple) else loss[0]))
            sd_a, sd_t = agent.actor.state_dict(), agent.actor_target.state_dict()
            for k in sd_a:
                assert sd_a[k].shape == sd_t[k].shape, (method, k)
        clone = agent.clone(1)
        x = torch.rand(4, 3, 16, 16) * 255
        torch.testing.assert_close(clone.actor(x), agent.actor(x))

    def test_ddpg_multi_input_mutations(self):
        from agilerl_amd.algorithms.ddpg import DDPG
        from agilerl_amd.spaces import Box, DictSpace

        torch.manual_seed(1)
        space = DictSpace({"vec": Box(-1, 1, (5,)), "img": Box(0, 255, (1, 8, 8))})
        agent = DDPG(space, Box(-1, 1, (2,)), batch_size=8,
                     net_config={"arch": "multi_input"})

        def obs(n):
            return {"vec": torch.randn(n, 5), "img": torch.rand(n, 1, 8, 8) * 255}

        batch = {
            "obs": obs(8), "action": torch.rand(8, 2) * 2 - 1,
            "reward": torch.randn(8, 1), "next_obs": obs(8),
            "done": torch.zeros(8, 1),
        }
        for method in [m for m in agent.mutation_methods if "activation" not in m]:
            agent.apply_architecture_mutation(method)
            out = agent.learn(batch)
            vals = out if isinstance(out, tuple) else (out,)
            assert all(np.isfinite(float(v)) for v in vals if v is not None)
            for name, tgt in (("actor", "actor_target"), ("critic", "critic_target")):
                sd_a = getattr(agent, name).state_dict()
                sd_t = getattr(agent, tgt).state_dict()
                for k in sd_a:
                    assert sd_a[k].shape == sd_t[k].shape, (name, method, k)
        clone = agent.clone(2)
        o = obs(4)
        torch.testing.assert_close(clone.actor(o), agent.actor(o))
