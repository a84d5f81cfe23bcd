"""Multi-process distributed tests (gloo backend, CPU, world_size=2)."""

import os

import numpy as np
import torch
import torch.multiprocessing as mp


def _find_free_port():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _pop_worker(rank, world_size, port, results):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)

    from agilerl_amd.algorithms.dqn import DQN
    from agilerl_amd.hpo import Mutations, TournamentSelection
    from agilerl_amd.parallel import DistributedPopulation, DistributedState
    from agilerl_amd.spaces import Box, Discrete

    DistributedState.reset()
    state = DistributedState.get()
    assert state.world_size == world_size

    def factory(index):
        torch.manual_seed(100 + index)
        return DQN(Box(-1, 1, (4,)), Discrete(2), index=index)

    pop = DistributedPopulation(factory, pop_size=4)
    assert len(pop.local_indices) == 2

    # give slot (rank 1's slot 3) the best fitness
    for slot in pop.local_indices:
        pop.agents[slot].fitness.append(float(slot))

    fits = pop.gather_fitness()
    np.testing.assert_allclose(fits, [0.0, 1.0, 2.0, 3.0])

    tour = TournamentSelection(tournament_size=4, elitism=True, rng=np.random.default_rng(0))
    muts = Mutations(no_mutation=1.0, architecture=0, parameters=0, activation=0, rl_hp=0)
    pop.evolve(tour, muts)

    # elite (parent slot 3) must now occupy slot 0 on rank 0 with identical weights
    x = torch.randn(3, 4, generator=torch.Generator().manual_seed(5))
    if rank == 0:
        q_elite = pop.agents[0].actor(x).detach()
        results[0] = q_elite.numpy()
    torch.distributed.barrier()
    torch.distributed.destroy_process_group()


def _expected_elite_q():
    from agilerl_amd.algorithms.dqn import DQN
    from agilerl_amd.spaces import Box, Discrete

    torch.manual_seed(103)
    agent = DQN(Box(-1, 1, (4,)), Discrete(2), index=3)
    x = torch.randn(3, 4, generator=torch.Generator().manual_seed(5))
    return agent.actor(x).detach().numpy()


def test_distributed_population_gloo():
    port = _find_free_port()
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [
            ctx.Process(target=_pop_worker, args=(r, 2, port, results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=180)
            assert p.exitcode == 0
        q_elite = np.array(results[0])
    np.testing.assert_allclose(q_elite, _expected_elite_q(), rtol=1e-5, atol=1e-6)


def _ddp_worker(rank, world_size, port):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)

    from agilerl_amd.parallel import DistributedState, allreduce_gradients, broadcast_module, wrap_ddp

    DistributedState.reset()
    DistributedState.get()

    torch.manual_seed(0)
    net = torch.nn.Linear(8, 4)
    broadcast_module(net, src=0)
    wrap_ddp(net)

    torch.manual_seed(rank)  # different data per rank
    x = torch.randn(16, 8)
    loss = net(x).pow(2).mean()
    loss.backward()
    allreduce_gradients(net)

    # gradients must be identical across ranks after allreduce
    g = net.weight.grad.clone()
    gather = [torch.zeros_like(g) for _ in range(world_size)]
    torch.distributed.all_gather(gather, g)
    assert torch.allclose(gather[0], gather[1], atol=1e-6)
    torch.distributed.destroy_process_group()


def test_ddp_allreduce_gloo():
    port = _find_free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_ddp_worker, args=(r, 2, port)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0


def _dist_train_worker(rank, world_size, port, results):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)

    import numpy as np

    from agilerl_amd.algorithms.ppo import PPO
    from agilerl_amd.envs import CartPoleVecEnv
    from agilerl_amd.hpo import Mutations, TournamentSelection
    from agilerl_amd.parallel import DistributedState
    from agilerl_amd.training.train_distributed import train_on_policy_distributed

    DistributedState.reset()
    DistributedState.get()

    def agent_factory(slot):
        torch.manual_seed(slot)
        env = CartPoleVecEnv(1)
        return PPO(env.single_observation_space, env.single_action_space,
                   index=slot, learn_step=32, batch_size=64,
                   net_config={"arch": "mlp", "hidden_size": [32]})

    def env_factory(slot):
        return CartPoleVecEnv(8, seed=slot)

    agents, hist = train_on_policy_distributed(
        agent_factory, env_factory, pop_size=4,
        max_steps=600, evo_steps=256,
        tournament=TournamentSelection(2, True, rng=np.random.default_rng(0)),
        mutation=Mutations(no_mutation=0.6, architecture=0.0, parameters=0.0,
                           activation=0.0, rl_hp=0.4, rand_seed=rank),
        use_graph=False, verbose=False,
    )
    results[rank] = (sorted(agents.keys()), len(hist))
    torch.distributed.destroy_process_group()


def test_train_on_policy_distributed_gloo():
    port = _find_free_port()
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [ctx.Process(target=_dist_train_worker, args=(r, 2, port, results))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
            assert p.exitcode == 0
        # rank 0 owns even slots, rank 1 odd slots
        assert results[0][0] == [0, 2]
        assert results[1][0] == [1, 3]
        assert results[0][1] >= 1


def test_adopt_agent_state():
    from agilerl_amd.algorithms.ppo import PPO
    from agilerl_amd.parallel.population_runtime import adopt_agent_state
    from agilerl_amd.spaces import Box, Discrete

    torch.manual_seed(0)
    a = PPO(Box(-1, 1, (4,)), Discrete(2))
    b = a.clone(1)
    with torch.no_grad():
        for p in b.actor.parameters():
            p.add_(1.0)
    b.fitness = [5.0]
    assert adopt_agent_state(a, b)
    x = torch.randn(3, 4)
    assert torch.allclose(a.actor(x), b.actor(x))
    assert a.fitness == [5.0]
    # architecture mismatch refuses
    c = a.clone(2)
    c.apply_architecture_mutation("encoder.add_node", numb_new_nodes=16)
    assert not adopt_agent_state(a, c)
    # lr change adopts but invalidates captured graphs
    d = a.clone(3)
    d.lr = 1e-5
    a._learn_graph = object()
    assert adopt_agent_state(a, d)
    assert a._learn_graph is None and a.lr == 1e-5


def _nonfinite_worker(rank, world_size, port, results):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)

    from agilerl_amd.parallel import DistributedState

    DistributedState.reset()
    DistributedState.get()

    class _Carrier:
        device = "cpu"
        from agilerl_amd.algorithms.llm.base import LLMAlgorithm

        raise_if_loss_not_finite_on_any_rank = LLMAlgorithm.raise_if_loss_not_finite_on_any_rank

    carrier = _Carrier()
    # only rank 1's local loss is NaN, but BOTH ranks must raise
    loss = torch.tensor(float("nan") if rank == 1 else 0.5)
    try:
        carrier.raise_if_loss_not_finite_on_any_rank(loss)
        results[rank] = "no_raise"
    except RuntimeError:
        results[rank] = "raised"
    torch.distributed.barrier()
    torch.distributed.destroy_process_group()


def test_nonfinite_loss_aborts_all_ranks():
    """coordinated abort — a NaN on one rank raises on every
    rank, so no healthy rank hangs in the next collective."""
    ctx = mp.get_context("spawn")
    results = ctx.Manager().dict()
    port = _find_free_port()
    procs = [ctx.Process(target=_nonfinite_worker, args=(r, 2, port, results)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(120)
        assert p.exitcode == 0
    assert results[0] == "raised" and results[1] == "raised"


def _seqlen_worker(rank, world_size, port, results):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)

    from agilerl_amd.parallel import DistributedState

    DistributedState.reset()
    DistributedState.get()

    class _Carrier:
        device = "cpu"
        from agilerl_amd.algorithms.llm.base import LLMAlgorithm

        check_seq_len_agreement = LLMAlgorithm.check_seq_len_agreement

    carrier = _Carrier()
    # agreeing lengths pass on both ranks
    carrier.check_seq_len_agreement(128)
    # mismatched lengths raise on both ranks
    try:
        carrier.check_seq_len_agreement(128 + rank)
        results[rank] = "no_raise"
    except RuntimeError as e:
        results[rank] = "raised" if "mismatch" in str(e) else f"wrong: {e}"
    torch.distributed.barrier()
    torch.distributed.destroy_process_group()


def test_seq_len_mismatch_raises_all_ranks():
    ctx = mp.get_context("spawn")
    results = ctx.Manager().dict()
    port = _find_free_port()
    procs = [ctx.Process(target=_seqlen_worker, args=(r, 2, port, results)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(120)
        assert p.exitcode == 0
    assert results[0] == "raised" and results[1] == "raised"


def _metrics_shard_worker(rank, world_size, port, results):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)

    from agilerl_amd.parallel import DistributedState, aggregate_metrics_across_ranks

    DistributedState.reset()
    DistributedState.get()

    # cross-rank metric mean
    stats = aggregate_metrics_across_ranks({"loss": float(rank), "note": "x"})
    results[f"loss_{rank}"] = stats["loss"]
    results[f"note_{rank}"] = stats["note"]

    # per-rank prompt sharding: disjoint slices
    class _Tok:
        pad_token_id = 0
        eos_token = "<eos>"
        pad_token = "<pad>"
        chat_template = None

        def __call__(self, texts, **kw):
            import torch as _t

            ids = _t.ones(len(texts), 4, dtype=_t.long)
            return {"input_ids": ids, "attention_mask": _t.ones_like(ids)}

    from agilerl_amd.llm_envs import ReasoningGym

    gym = ReasoningGym([f"p{i}" for i in range(10)], list(range(10)),
                       lambda c, a: 0.0, _Tok(), data_batch_size=2)
    results[f"prompts_{rank}"] = tuple(gym.prompts)
    torch.distributed.barrier()
    torch.distributed.destroy_process_group()


def test_metric_aggregation_and_data_sharding():
    ctx = mp.get_context("spawn")
    results = ctx.Manager().dict()
    port = _find_free_port()
    procs = [ctx.Process(target=_metrics_shard_worker, args=(r, 2, port, results)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(120)
        assert p.exitcode == 0
    assert results["loss_0"] == results["loss_1"] == 0.5  # mean of 0 and 1
    assert results["note_0"] == "x"
    assert set(results["prompts_0"]).isdisjoint(results["prompts_1"])
    assert len(results["prompts_0"]) + len(results["prompts_1"]) == 10


def _wrap_models_worker(rank, world_size, port):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)

    from agilerl_amd.algorithms.dqn import DQN
    from agilerl_amd.parallel import DistributedState, broadcast_module
    from agilerl_amd.spaces import Box, Discrete

    DistributedState.reset()
    DistributedState.get()

    torch.manual_seed(0)
    agent = DQN(Box(-1, 1, (4,)), Discrete(2), batch_size=8)
    broadcast_module(agent.actor, src=0)
    broadcast_module(agent.actor_target, src=0)
    agent.wrap_models()

    # DIFFERENT experiences per rank: without gradient sync the ranks diverge
    g = torch.Generator().manual_seed(rank + 1)
    batch = {
        "obs": torch.randn(8, 4, generator=g),
        "action": torch.randint(0, 2, (8, 1), generator=g),
        "reward": torch.randn(8, 1, generator=g),
        "next_obs": torch.randn(8, 4, generator=g),
        "done": torch.zeros(8, 1),
    }
    for _ in range(3):
        agent.learn(batch)

    # parameters must be bitwise-identical across ranks after synced steps
    for p in agent.actor.parameters():
        gather = [torch.zeros_like(p.data) for _ in range(world_size)]
        torch.distributed.all_gather(gather, p.data)
        assert torch.allclose(gather[0], gather[1], atol=1e-7), "ranks diverged"
    torch.distributed.destroy_process_group()


def test_wrap_models_learn_syncs_gradients_gloo():
    """wrap_models attached bucketer hooks but no RL learn path
    drained them before optimizer.step() — ranks silently diverged.  The
    OptimizerWrapper now finalizes bucketers inside step()."""
    port = _find_free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_wrap_models_worker, args=(r, 2, port)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
