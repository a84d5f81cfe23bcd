#!/usr/bin/env python3
"""Headline benchmark: PPO LunarLander pop=8 evolutionary HPO.

BASELINE.json metric: env steps/sec (whole node) for PPO LunarLander
pop=8 at 1/2/4/8 GPUs (one agent per GPU at N=8; agents round-robin on
fewer GPUs).  One bench "step" = every population agent runs one
collect(learn_step x num_envs env-steps) + PPO-update cycle; an
evolution round (fitness all-gather + rank-0 tournament plan broadcast +
winner weight transfer + mutation, all over RCCL/xGMI) fires every
EVO_EVERY steps INSIDE the timed region.

Launch (the driver does this for N>1):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port P bench.py --gpus N --steps K --warmup W

Single line of JSON on rank 0 at the end (driver contract).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from agilerl_amd.algorithms.ppo import PPO  # noqa: E402
from agilerl_amd.components.rollout_buffer import RolloutBuffer  # noqa: E402
from agilerl_amd.envs import LunarLanderVecEnv  # noqa: E402
from agilerl_amd.hpo import Mutations, TournamentSelection  # noqa: E402
from agilerl_amd.parallel import DistributedPopulation, DistributedState, barrier  # noqa: E402
from agilerl_amd.rollouts.on_policy import collect_rollouts, collect_rollouts_device  # noqa: E402

POP_SIZE = 8
NUM_ENVS = 64
LEARN_STEP = 128  # rollout length per cycle
EVO_EVERY = 4


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=8)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument(
        "--num-envs", type=int, default=None,
        help="envs per agent (default: 65536 on GPU via the hipGraph collector, 64 on CPU)",
    )
    p.add_argument("--learn-step", type=int, default=LEARN_STEP)
    p.add_argument("--pop-size", type=int, default=POP_SIZE)
    p.add_argument(
        "--workload",
        choices=["ppo", "grpo", "grpo-rollout", "dqn", "rainbow", "maddpg"], default="ppo",
        help="BASELINE configs: ppo = headline (LunarLander pop=8); "
        "dqn = config 1 (CartPole pop=1 CPU plumbing); rainbow = config 3 "
        "(PER+noisy CNN on the Pong-like visual env, pop=4); maddpg = "
        "config 4 (speaker_listener pop=8); grpo = config 5 (Llama-3-8B "
        "random-init, DP, synthetic tokens)",
    )
    p.add_argument("--seq-len", type=int, default=1024, help="grpo: prompt+completion length")
    p.add_argument("--grpo-batch", type=int, default=32, help="grpo: sequences per step per rank")
    p.add_argument("--model-size", choices=["8b", "tiny"], default="8b")
    p.add_argument("--no-graph", action="store_true", help="disable hipGraph collector")
    p.add_argument("--packing", action="store_true", help="grpo: padding-free packed logprob passes")
    p.add_argument("--micro-batch", type=int, default=None, help="grpo: micro batch size per backward")
    p.add_argument("--grad-ckpt", action="store_true",
                   help="grpo: enable gradient checkpointing (default OFF: 288 GB "
                   "fits 8B seq-1k activations at micro-batch 16; measured +36%%)")
    p.add_argument("--no-grad-ckpt", action="store_true", help=argparse.SUPPRESS)
    return p.parse_args()


class BenchRunner:
    def __init__(self, args):
        self.args = args
        self.state = DistributedState.get()
        self.device = self.state.device
        seed = 1234 + self.state.rank
        np.random.seed(seed)
        torch.manual_seed(seed)

        # minibatch sized to the rollout so big env counts keep the GPU in
        # few large GEMMs instead of hundreds of small update steps
        mb_size = max(2048, args.num_envs * args.learn_step // 16)
        # HP-mutation bounds sized to THIS config: the generic defaults cap
        # batch_size at 4096, so one rl_hp mutation would shred the 524k
        # minibatch into 128x more update steps and the slowdown spreads
        # through clones (observed: +7 s/step per evolution round)
        from agilerl_amd.algorithms.core.registry import HyperparameterConfig, RLParameter

        hp = HyperparameterConfig(
            lr=RLParameter(min=1e-5, max=1e-2),
            batch_size=RLParameter(min=mb_size // 4, max=mb_size * 2, dtype=int),
            clip_coef=RLParameter(min=0.05, max=0.4),
            ent_coef=RLParameter(min=1e-4, max=0.05),
            update_epochs=RLParameter(min=1, max=4, dtype=int),
        )

        probe_env = LunarLanderVecEnv(1)
        obs_space, act_space = probe_env.single_observation_space, probe_env.single_action_space

        def factory(index: int) -> PPO:
            return PPO(
                observation_space=obs_space,
                action_space=act_space,
                index=index,
                hp_config=hp,
                learn_step=args.learn_step,
                batch_size=mb_size,
                lr=3e-4,
                update_epochs=4,
                net_config={"arch": "mlp", "hidden_size": [64, 64]},
                device=self.device,
            )

        self.pop = DistributedPopulation(factory, args.pop_size)
        self.tournament = TournamentSelection(tournament_size=2, elitism=True,
                                              rng=np.random.default_rng(7))
        self.mutations = Mutations(
            no_mutation=0.4, architecture=0.2, parameters=0.1, activation=0.1,
            rl_hp=0.2, rand_seed=seed, device=self.device,
        )
        # per-slot envs + rollout buffers + carried state
        self.envs = {}
        self.buffers = {}
        self.carried = {}
        self.fit_window = {}
        self.collectors = {}
        self.use_graph = torch.cuda.is_available() and not args.no_graph
        for slot in self.pop.local_indices:
            self._init_slot(slot)
        self.step_count = 0

    def _init_slot(self, slot: int) -> None:
        if torch.cuda.is_available():
            # device-resident env: the collect loop never leaves HBM
            from agilerl_amd.envs.torch_envs import LunarLanderTorchVecEnv

            self.envs[slot] = LunarLanderTorchVecEnv(
                self.args.num_envs, device=self.device, seed=1000 + slot
            )
        else:
            self.envs[slot] = LunarLanderVecEnv(self.args.num_envs, seed=1000 + slot)
        agent = self.pop.agents[slot]
        self.buffers[slot] = RolloutBuffer(
            capacity=agent.learn_step, num_envs=self.args.num_envs,
            device=self.device, gamma=agent.gamma, gae_lambda=agent.gae_lambda,
        )
        self.carried[slot] = (None, None)
        self.fit_window[slot] = []
        self.collectors.pop(slot, None)
        if self.use_graph:
            from agilerl_amd.rollouts.graph_collector import GraphedPPOCollector

            self.collectors[slot] = GraphedPPOCollector(agent, self.envs[slot], agent.learn_step)

    def bench_step(self) -> int:
        """One population cycle; returns env steps consumed (this rank)."""
        steps = 0
        for slot in self.pop.local_indices:
            agent = self.pop.agents[slot]
            buffer = self.buffers[slot]
            if buffer.capacity != agent.learn_step:
                buffer = RolloutBuffer(
                    capacity=agent.learn_step, num_envs=self.args.num_envs,
                    device=self.device, gamma=agent.gamma, gae_lambda=agent.gae_lambda,
                )
                self.buffers[slot] = buffer
            env = self.envs[slot]
            if self.use_graph:
                collector = self.collectors.get(slot)
                if collector is None or collector.agent is not agent or collector.n_steps != agent.learn_step:
                    from agilerl_amd.rollouts.graph_collector import GraphedPPOCollector

                    collector = GraphedPPOCollector(agent, env, agent.learn_step)
                    self.collectors[slot] = collector
                flat, stats = collector.collect()
                agent.learn(flat)
            else:
                obs, done = self.carried[slot]
                collect = (
                    collect_rollouts_device if getattr(env, "is_torch", False) else collect_rollouts
                )
                obs, done, stats = collect(agent, env, buffer, agent.learn_step, obs, done)
                self.carried[slot] = (obs, done)
                agent.learn(buffer)
            n = agent.learn_step * self.args.num_envs
            agent.steps[-1] += n
            steps += n
            if "mean_episode_return" in stats:
                self.fit_window[slot].append(stats["mean_episode_return"])
                self.fit_window[slot] = self.fit_window[slot][-5:]
                agent.fitness.append(float(np.mean(self.fit_window[slot])))
        self.step_count += 1
        if self.step_count % EVO_EVERY == 0:
            from agilerl_amd.parallel.population_runtime import adopt_agent_state

            old_agents = dict(self.pop.agents)
            self.pop.evolve(self.tournament, self.mutations)
            for slot in self.pop.local_indices:
                old, new = old_agents.get(slot), self.pop.agents[slot]
                if old is not None and old is not new and adopt_agent_state(old, new):
                    self.pop.agents[slot] = old  # captured graphs stay valid
                    continue
                if slot not in self.envs:
                    self._init_slot(slot)
                else:
                    self.carried[slot] = (None, None)
        return steps


LLAMA3_8B = dict(
    model_type="llama", vocab_size=128256, hidden_size=4096, intermediate_size=14336,
    num_hidden_layers=32, num_attention_heads=32, num_key_value_heads=8,
    max_position_embeddings=8192, rope_theta=500000.0, pad_token_id=0,
)
LLAMA_TINY = dict(
    model_type="llama", vocab_size=2048, hidden_size=256, intermediate_size=512,
    num_hidden_layers=4, num_attention_heads=8, num_key_value_heads=4,
    max_position_embeddings=2048, pad_token_id=0,
)


class GrpoBenchRunner:
    """BASELINE config 5: GRPO on random-init Llama-3-8B, DP over all ranks,
    synthetic prompt/response tokens (generation is not timed — the config
    names synthetic response tokens), group-advantage + fused logprob/loss
    HIP kernels, adapter-grad RCCL all-reduce over xGMI."""

    def __init__(self, args):
        from agilerl_amd.algorithms.llm.grpo import GRPO

        self.args = args
        self.state = DistributedState.get()
        self.device = self.state.device
        torch.manual_seed(1234)  # identical base weights on every rank
        cfg = LLAMA3_8B if args.model_size == "8b" else LLAMA_TINY
        dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32
        self.agent = GRPO(
            model_config=dict(cfg),
            dtype=dtype,
            lora_config={"r": 16, "lora_alpha": 32},
            group_size=min(8, max(1, args.grpo_batch)),
            micro_batch_size=args.micro_batch or (16 if args.model_size == "8b" else 8),
            update_epochs=1,
            beta=0.04,
            lr=5e-6,
            gradient_checkpointing=bool(args.grad_ckpt),
            use_packing=args.packing,
            device=self.device,
        )
        self.vocab = cfg["vocab_size"]
        np.random.seed(77 + self.state.rank)  # rank-sharded data

    def _synthetic_batch(self):
        B, T = self.args.grpo_batch, self.args.seq_len
        ids = torch.from_numpy(np.random.randint(1, self.vocab, (B, T))).to(self.device)
        P = T // 2
        pos = torch.arange(T - 1, device=self.device).unsqueeze(0)
        action_mask = (pos + 1 >= P).float().expand(B, T - 1)
        rewards = torch.from_numpy(np.random.rand(B).astype(np.float32))
        return {"ids": ids, "action_mask": action_mask, "rewards": rewards}

    def bench_step(self) -> int:
        batch = self._synthetic_batch()
        self.agent.learn(batch)
        return int(batch["ids"].numel())


class GrpoRolloutBenchRunner(GrpoBenchRunner):
    """grpo-rollout: generation INSIDE the timed region — the dominant RFT
    cost the reference covers with colocated vLLM.  Each step: paged-engine
    generation of C completion tokens per sequence, then the GRPO learn
    phase on the generated batch."""

    def __init__(self, args):
        super().__init__(args)
        self.agent.generation = "paged"
        self.prompt_len = max(8, args.seq_len // 2)
        self.completion_len = args.seq_len - self.prompt_len
        self.agent.max_completion_tokens = self.completion_len
        self.gen_tokens = 0
        self.gen_seconds = 0.0

    def bench_step(self) -> int:
        import sys as _sys
        import time as _t

        B, P = self.args.grpo_batch, self.prompt_len
        ids = torch.from_numpy(np.random.randint(1, self.vocab, (B, P))).to(self.device)
        mask = torch.ones_like(ids)
        t0 = _t.perf_counter()
        seqs = self.agent.get_action({"input_ids": ids, "attention_mask": mask})
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self.gen_seconds += _t.perf_counter() - t0
        self.gen_tokens += B * self.completion_len
        eng = getattr(self.agent, "_decode_engine", None)
        if eng is not None and not getattr(self, "_graph_reported", False):
            self._graph_reported = True
            graphs = {k: d.graph is not None for k, d in eng._graph_decoders.items()}
            print(f"[bench] decode graphs: {graphs} use_graph={eng._use_decode_graph}",
                  file=_sys.stderr, flush=True)
        T = seqs.shape[1]
        pos = torch.arange(T - 1, device=self.device).unsqueeze(0)
        action_mask = (pos + 1 >= P).float().expand(B, T - 1)
        rewards = torch.from_numpy(np.random.rand(B).astype(np.float32))
        self.agent.learn({"ids": seqs, "action_mask": action_mask, "rewards": rewards})
        return int(seqs.numel())


def run_grpo(args, rollout: bool = False):
    runner = GrpoRolloutBenchRunner(args) if rollout else GrpoBenchRunner(args)
    state = runner.state
    use_cuda = torch.cuda.is_available()
    for _ in range(args.warmup):
        runner.bench_step()
    if rollout:  # generation stats restart with the timed region
        runner.gen_tokens = 0
        runner.gen_seconds = 0.0
    barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    local_tokens = 0
    for _ in range(args.steps):
        local_tokens += runner.bench_step()
    if use_cuda:
        torch.cuda.synchronize()
    barrier()
    elapsed = time.perf_counter() - t0
    if state.is_distributed:
        import torch.distributed as dist

        t = torch.tensor([elapsed, float(local_tokens)])
        if state.backend == "nccl":
            t = t.to(state.device)
        dist.all_reduce(t[0:1], op=dist.ReduceOp.MAX)
        dist.all_reduce(t[1:2], op=dist.ReduceOp.SUM)
        elapsed, total_tokens = float(t[0]), float(t[1])
    else:
        total_tokens = float(local_tokens)
    if state.is_main:
        config = {
            "model": "Llama-3-8B (random init)" if args.model_size == "8b" else "llama-tiny",
            "global_batch": args.grpo_batch * state.world_size,
            "seq_len": args.seq_len,
            "parallelism": f"dp{state.world_size} (adapter-grad RCCL allreduce)",
            "algo": "GRPO group_size=8 beta=0.04 LoRA r=16",
        }
        if rollout:
            config["phase"] = "generate(paged engine)+learn end-to-end"
            if runner.gen_seconds > 0:
                config["generation_tokens_per_sec"] = runner.gen_tokens / runner.gen_seconds
                config["generation_fraction_of_step"] = runner.gen_seconds / elapsed
        result = {
            "metric": "rollout_tokens_per_sec" if rollout else "train_tokens_per_sec",
            "value": total_tokens / elapsed,
            "unit": "tokens/s",
            "n_gpus": state.world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_cuda else "fp32",
            "data": "synthetic prompt/response tokens, random-init weights",
            "config": config,
        }
        print(json.dumps(result), flush=True)
    if state.is_distributed:
        import torch.distributed as dist

        dist.destroy_process_group()


def _emit(state, metric, value, unit, args, elapsed, extra_config, dtype="fp32",
          scaling="strong", data="synthetic"):
    if not state.is_main:
        return
    print(json.dumps({
        "metric": metric, "value": value, "unit": unit,
        "n_gpus": state.world_size, "steps": args.steps, "warmup": args.warmup,
        "ms_per_step": elapsed / args.steps * 1000.0, "higher_is_better": True,
        "scaling": scaling, "vs_baseline": None, "dtype": dtype, "data": data,
        "config": extra_config,
    }), flush=True)


def _timed_loop(state, step_fn, args):
    use_cuda = torch.cuda.is_available()
    for _ in range(args.warmup):
        step_fn()
    barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    local = 0
    for _ in range(args.steps):
        local += step_fn()
    if use_cuda:
        torch.cuda.synchronize()
    barrier()
    elapsed = time.perf_counter() - t0
    if state.is_distributed:
        import torch.distributed as dist

        t = torch.tensor([elapsed, float(local)])
        if state.backend == "nccl":
            t = t.to(state.device)
        dist.all_reduce(t[0:1], op=dist.ReduceOp.MAX)
        dist.all_reduce(t[1:2], op=dist.ReduceOp.SUM)
        elapsed, total = float(t[0]), float(t[1])
    else:
        total = float(local)
    return elapsed, total


def run_dqn(args):
    """BASELINE config 1: DQN CartPole pop=1 on CPU vectorized envs."""
    from agilerl_amd.algorithms import DQN
    from agilerl_amd.components import ReplayBuffer
    from agilerl_amd.envs import CartPoleVecEnv

    state = DistributedState.get()
    num_envs = min(args.num_envs or 64, 256)
    env = CartPoleVecEnv(num_envs, seed=0)
    agent = DQN(env.observation_space, env.action_space, batch_size=128, lr=1e-3,
                device="cpu")
    memory = ReplayBuffer(100_000)
    obs, _ = env.reset()
    eps = [1.0]

    def step_fn():
        nonlocal obs
        steps = 0
        for it in range(128):
            action = agent.get_action(obs, epsilon=eps[0])
            next_obs, reward, term, trunc, info = env.step(action)
            memory.add(obs=obs, action=action, reward=reward, next_obs=next_obs,
                       done=term.astype(np.float32))
            obs = next_obs
            eps[0] = max(0.05, eps[0] * 0.999)
            steps += num_envs
            if len(memory) >= 256 and it % agent.learn_step == 0:
                agent.learn(memory.sample(agent.batch_size))
        return steps

    elapsed, total = _timed_loop(state, step_fn, args)
    _emit(state, "env_steps_per_sec", total / elapsed, "steps/s", args, elapsed,
          {"model": "DQN mlp[64,64]", "env": "CartPole-v1 (first-party)",
           "global_batch": 128, "num_envs": num_envs, "pop_size": 1,
           "parallelism": "cpu single-agent (plumbing check)"})


def run_rainbow(args):
    """BASELINE config 3 shape: Rainbow-DQN (PER + n-step + noisy + C51 HIP
    kernels) on the Pong-like visual env, population across ranks."""
    from agilerl_amd.algorithms import RainbowDQN
    from agilerl_amd.components import PrioritizedReplayBuffer
    from agilerl_amd.envs import CatchPongVecEnv

    state = DistributedState.get()
    device = state.device
    pop_size = 4
    num_envs = args.num_envs or (2048 if torch.cuda.is_available() else 8)

    def factory(index):
        return RainbowDQN(
            CatchPongVecEnv(1).single_observation_space,
            CatchPongVecEnv(1).single_action_space,
            index=index, batch_size=256, lr=1e-4, n_step=3,
            net_config={"arch": "cnn", "channel_size": [32, 64, 64],
                        "kernel_size": [8, 4, 3], "stride_size": [4, 2, 1]},
            device=device,
        )

    pop = DistributedPopulation(factory, pop_size)
    use_torch_env = torch.cuda.is_available()
    if use_torch_env:
        from agilerl_amd.envs.torch_envs import CatchPongTorchVecEnv

        envs = {s: CatchPongTorchVecEnv(num_envs, device=device, seed=100 + s)
                for s in pop.local_indices}
        storage = device  # replay lives in HBM (288 GB): no PCIe in the loop
    else:
        envs = {s: CatchPongVecEnv(num_envs, seed=100 + s) for s in pop.local_indices}
        storage = None
    mems = {
        s: PrioritizedReplayBuffer(100_000, n_step=3, gamma=0.99, device=device,
                                   storage_device=storage)
        for s in pop.local_indices
    }
    obs_map = {s: envs[s].reset()[0] for s in pop.local_indices}

    def step_fn():
        steps = 0
        for slot in pop.local_indices:
            agent, env, mem = pop.agents[slot], envs[slot], mems[slot]
            obs = obs_map[slot]
            for it in range(32):
                action = agent.get_action(obs)
                next_obs, reward, term, trunc, info = env.step(action)
                done = term.float() if use_torch_env else term.astype(np.float32)
                mem.add(obs=obs, action=action, reward=reward, next_obs=next_obs,
                        done=done)
                obs = next_obs
                steps += num_envs
                if len(mem) >= 2000 and it % agent.learn_step == 0:
                    batch = mem.sample(agent.batch_size, beta=0.5)
                    agent.learn(batch)
                    mem.update_priorities(batch["idxs"], agent.last_td_errors)
            obs_map[slot] = obs
        return steps

    elapsed, total = _timed_loop(state, step_fn, args)
    _emit(state, "env_steps_per_sec", total / elapsed, "steps/s", args, elapsed,
          {"model": "RainbowDQN cnn[32,64,64] C51 PER n-step3 noisy",
           "env": "CatchPong-v0 (first-party Pong-like, 4x84x84 uint8)",
           "global_batch": 256, "num_envs_per_agent": num_envs, "pop_size": pop_size,
           "parallelism": f"population-parallel dp{state.world_size}"})


def run_maddpg(args):
    """BASELINE config 4: MADDPG speaker_listener pop=8, multi-agent replay."""
    from agilerl_amd.algorithms import MADDPG
    from agilerl_amd.components import ReplayBuffer
    from agilerl_amd.envs.mpe import SpeakerListenerVecEnv

    state = DistributedState.get()
    device = state.device
    pop_size = 8
    num_envs = args.num_envs or (2048 if torch.cuda.is_available() else 64)
    probe = SpeakerListenerVecEnv(1)

    def factory(index):
        return MADDPG(
            probe.observation_spaces, probe.action_spaces, agent_ids=probe.agents,
            index=index, batch_size=512, device=device,
            net_config={"arch": "mlp", "hidden_size": [64, 64]},
        )

    pop = DistributedPopulation(factory, pop_size)
    use_torch_env = torch.cuda.is_available()
    if use_torch_env:
        from agilerl_amd.envs.torch_mpe import SpeakerListenerTorchVecEnv

        envs = {s: SpeakerListenerTorchVecEnv(num_envs, device=device, seed=100 + s)
                for s in pop.local_indices}
        storage = device
    else:
        envs = {s: SpeakerListenerVecEnv(num_envs, seed=100 + s) for s in pop.local_indices}
        storage = None
    mems = {s: ReplayBuffer(100_000, device=device, storage_device=storage)
            for s in pop.local_indices}
    obs_map = {s: envs[s].reset()[0] for s in pop.local_indices}
    agent_ids = probe.agents

    def step_fn():
        steps = 0
        for slot in pop.local_indices:
            agent, env, mem = pop.agents[slot], envs[slot], mems[slot]
            obs = obs_map[slot]
            for it in range(32):
                env_actions, raw = agent.get_action(obs)
                next_obs, rewards, term, trunc, info = env.step(env_actions)
                if use_torch_env:
                    done = {a: term[a].float() for a in agent_ids}
                else:
                    done = {a: term[a].astype(np.float32) for a in agent_ids}
                mem.add(obs=obs, action=raw,
                        reward={a: rewards[a] for a in agent_ids},
                        next_obs=next_obs, done=done)
                obs = next_obs
                steps += num_envs
                if len(mem) >= 2000 and it % agent.learn_step == 0:
                    agent.learn(mem.sample(agent.batch_size))
            obs_map[slot] = obs
        return steps

    elapsed, total = _timed_loop(state, step_fn, args)
    _emit(state, "env_steps_per_sec", total / elapsed, "steps/s", args, elapsed,
          {"model": "MADDPG mlp[64,64] centralized critics",
           "env": "simple_speaker_listener (first-party MPE)",
           "global_batch": 512, "num_envs_per_agent": num_envs, "pop_size": pop_size,
           "parallelism": f"population-parallel dp{state.world_size}"})


def main():
    args = parse_args()
    if args.num_envs is None and args.workload == "ppo":
        args.num_envs = 65536 if torch.cuda.is_available() else NUM_ENVS
    if args.workload == "grpo":
        return run_grpo(args)
    if args.workload == "grpo-rollout":
        return run_grpo(args, rollout=True)
    if args.workload == "dqn":
        return run_dqn(args)
    if args.workload == "rainbow":
        return run_rainbow(args)
    if args.workload == "maddpg":
        return run_maddpg(args)
    runner = BenchRunner(args)
    state = runner.state
    use_cuda = torch.cuda.is_available()

    for _ in range(args.warmup):
        runner.bench_step()

    barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    local_steps = 0
    for _ in range(args.steps):
        local_steps += runner.bench_step()
    if use_cuda:
        torch.cuda.synchronize()
    barrier()
    elapsed = time.perf_counter() - t0

    # max elapsed over ranks; total env steps over ranks
    if state.is_distributed:
        import torch.distributed as dist

        t = torch.tensor([elapsed, float(local_steps)])
        if state.backend == "nccl":
            t = t.to(state.device)
        dist.all_reduce(t[0:1], op=dist.ReduceOp.MAX)
        dist.all_reduce(t[1:2], op=dist.ReduceOp.SUM)
        elapsed = float(t[0])
        total_steps = float(t[1])
    else:
        total_steps = float(local_steps)

    if state.is_main:
        value = total_steps / elapsed
        result = {
            "metric": "env_steps_per_sec",
            "value": value,
            "unit": "steps/s",
            "n_gpus": state.world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic (first-party Box2D-free LunarLander dynamics, same obs/action spaces)",
            "config": {
                "model": "PPO mlp[64,64] evolutionary-HPO",
                "env": "LunarLander-v2 (first-party vectorized reimpl)",
                "global_batch": args.pop_size * args.num_envs * args.learn_step,
                "seq_len": args.learn_step,
                "num_envs_per_agent": args.num_envs,
                "pop_size": args.pop_size,
                "evo_every": EVO_EVERY,
                "parallelism": f"population-parallel dp{state.world_size} (one agent per GPU at 8)",
            },
        }
        print(json.dumps(result), flush=True)

    if state.is_distributed:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
