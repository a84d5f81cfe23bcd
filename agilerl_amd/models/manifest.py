"""Training manifest schema (pydantic v2).

Reference parity: ``agilerl/models/manifest.py:285`` (TrainingManifest with
sections algorithm / environment / network / mutation / replay_buffer /
selection_strategy / training), ``models/algo.py`` (AlgorithmSpec +
registry), ``models/training.py`` (TrainingSpec, ReplayBufferSpec),
``models/env.py`` (env specs), ``models/hpo.py`` (mutation/selection
specs), ``models/networks.py`` (net specs).  YAML manifests written for
the reference load unchanged for the shared fields.
"""

from __future__ import annotations

from typing import Any, Dict, List, Literal, Optional, Union

from pydantic import BaseModel, ConfigDict, Field, model_validator

__all__ = [
    "AlgorithmSpec",
    "GymEnvSpec",
    "PzEnvSpec",
    "LLMEnvSpec",
    "BanditEnvSpec",
    "OfflineEnvSpec",
    "NetworkSpec",
    "MutationSpec",
    "TournamentSelectionSpec",
    "ReplayBufferSpec",
    "TrainingSpec",
    "TrainingManifest",
    "ALGO_REGISTRY",
    "algo_workload",
]


class _Base(BaseModel):
    model_config = ConfigDict(extra="allow")


# ---------------------------------------------------------------------------
# Algorithm registry: name -> (import path, workload loop)
# ---------------------------------------------------------------------------
ALGO_REGISTRY: Dict[str, Dict[str, str]] = {
    "DQN": {"cls": "agilerl_amd.algorithms.dqn.DQN", "workload": "off_policy"},
    "RainbowDQN": {"cls": "agilerl_amd.algorithms.dqn_rainbow.RainbowDQN", "workload": "off_policy"},
    "Rainbow DQN": {"cls": "agilerl_amd.algorithms.dqn_rainbow.RainbowDQN", "workload": "off_policy"},
    "CQN": {"cls": "agilerl_amd.algorithms.cqn.CQN", "workload": "offline"},
    "DDPG": {"cls": "agilerl_amd.algorithms.ddpg.DDPG", "workload": "off_policy"},
    "TD3": {"cls": "agilerl_amd.algorithms.td3.TD3", "workload": "off_policy"},
    "PPO": {"cls": "agilerl_amd.algorithms.ppo.PPO", "workload": "on_policy"},
    "MADDPG": {"cls": "agilerl_amd.algorithms.maddpg.MADDPG", "workload": "multi_agent_off_policy"},
    "MATD3": {"cls": "agilerl_amd.algorithms.matd3.MATD3", "workload": "multi_agent_off_policy"},
    "IPPO": {"cls": "agilerl_amd.algorithms.ippo.IPPO", "workload": "multi_agent_on_policy"},
    "NeuralUCB": {"cls": "agilerl_amd.algorithms.neural_ucb.NeuralUCB", "workload": "bandit"},
    "NeuralTS": {"cls": "agilerl_amd.algorithms.neural_ts.NeuralTS", "workload": "bandit"},
    "GRPO": {"cls": "agilerl_amd.algorithms.llm.grpo.GRPO", "workload": "llm_reasoning"},
    "GSPO": {"cls": "agilerl_amd.algorithms.llm.gspo.GSPO", "workload": "llm_reasoning"},
    "CISPO": {"cls": "agilerl_amd.algorithms.llm.cispo.CISPO", "workload": "llm_reasoning"},
    "SFT": {"cls": "agilerl_amd.algorithms.llm.sft.SFT", "workload": "llm_sft"},
    "DPO": {"cls": "agilerl_amd.algorithms.llm.dpo.DPO", "workload": "llm_preference"},
    "PPOLLM": {"cls": "agilerl_amd.algorithms.llm.ppo_llm.PPOLLM", "workload": "llm_reasoning"},
    "LLM_PPO": {"cls": "agilerl_amd.algorithms.llm.ppo_llm.PPOLLM", "workload": "llm_reasoning"},
    "ReinforceLLM": {"cls": "agilerl_amd.algorithms.llm.reinforce_llm.ReinforceLLM", "workload": "llm_reasoning"},
    "LLM_REINFORCE": {"cls": "agilerl_amd.algorithms.llm.reinforce_llm.ReinforceLLM", "workload": "llm_reasoning"},
    # reference manifest spellings (configs/training/llm_finetuning/*.yaml)
    "LLMPPO": {"cls": "agilerl_amd.algorithms.llm.ppo_llm.PPOLLM", "workload": "llm_reasoning"},
    "LLMREINFORCE": {"cls": "agilerl_amd.algorithms.llm.reinforce_llm.ReinforceLLM", "workload": "llm_reasoning"},
}


def algo_workload(name: str) -> str:
    if name not in ALGO_REGISTRY:
        raise KeyError(f"Unknown algorithm '{name}'. Registered: {sorted(ALGO_REGISTRY)}")
    return ALGO_REGISTRY[name]["workload"]


def resolve_algo_class(name: str):
    import importlib

    if name not in ALGO_REGISTRY:
        raise KeyError(f"Unknown algorithm '{name}'. Registered: {sorted(ALGO_REGISTRY)}")
    path = ALGO_REGISTRY[name]["cls"]
    module, cls = path.rsplit(".", 1)
    return getattr(importlib.import_module(module), cls)


# ---------------------------------------------------------------------------
# Sections
# ---------------------------------------------------------------------------
class AlgorithmSpec(_Base):
    name: str
    hyperparameters: Dict[str, Any] = Field(default_factory=dict)


class GymEnvSpec(_Base):
    type: Literal["gym"] = "gym"
    env_id: str
    num_envs: int = 8
    env_kwargs: Dict[str, Any] = Field(default_factory=dict)


class PzEnvSpec(_Base):
    type: Literal["pettingzoo"] = "pettingzoo"
    env_id: str
    num_envs: int = 8
    continuous_actions: bool = False
    env_kwargs: Dict[str, Any] = Field(default_factory=dict)


class LLMEnvSpec(_Base):
    type: Literal["llm"] = "llm"
    env_type: str = "reasoning"  # reasoning | sft | preference | multiturn
    dataset: Optional[str] = None
    reward_fn: Optional[str] = None  # import path
    data_batch_size: int = 8
    group_size: int = 8
    max_prompt_tokens: int = 512
    max_completion_tokens: int = 512
    env_kwargs: Dict[str, Any] = Field(default_factory=dict)


class BanditEnvSpec(_Base):
    type: Literal["bandit"] = "bandit"
    env_id: Optional[str] = None
    context_dim: Optional[int] = None
    num_arms: Optional[int] = None
    env_kwargs: Dict[str, Any] = Field(default_factory=dict)


class OfflineEnvSpec(_Base):
    type: Literal["offline"] = "offline"
    env_id: str
    dataset_path: Optional[str] = None
    num_envs: int = 8


EnvSpec = Union[GymEnvSpec, PzEnvSpec, LLMEnvSpec, BanditEnvSpec, OfflineEnvSpec]


class NetworkSpec(_Base):
    arch: Optional[str] = None  # mlp | cnn | simba | lstm | multi_input
    encoder_config: Dict[str, Any] = Field(default_factory=dict)
    head_config: Dict[str, Any] = Field(default_factory=dict)
    latent_dim: int = 64


class MutationProbabilities(_Base):
    no_mutation: float = 0.2
    architecture: float = 0.2
    parameters: float = 0.2
    activation: float = 0.0
    rl_hp: float = 0.2

    @model_validator(mode="before")
    @classmethod
    def _reference_aliases(cls, data):
        # reference manifests spell these no_mut / arch_mut / params_mut /
        # act_mut / rl_hp_mut (configs/training/*); without the aliases
        # they validated but silently fell back to defaults
        if isinstance(data, dict):
            alias = {"no_mut": "no_mutation", "arch_mut": "architecture",
                     "params_mut": "parameters", "act_mut": "activation",
                     "rl_hp_mut": "rl_hp"}
            data = dict(data)
            for ref, ours in alias.items():
                if ref in data and ours not in data:
                    data[ours] = data.pop(ref)
        return data


class MutationSpec(_Base):
    probabilities: MutationProbabilities = Field(default_factory=MutationProbabilities)
    new_layer_prob: float = 0.2

    @model_validator(mode="before")
    @classmethod
    def _hoist_new_layer(cls, data):
        # the reference nests new_layer under probabilities
        if isinstance(data, dict):
            probs = data.get("probabilities")
            if isinstance(probs, dict) and "new_layer" in probs:
                data = dict(data)
                probs = dict(probs)
                data.setdefault("new_layer_prob", probs.pop("new_layer"))
                data["probabilities"] = probs
        return data
    mutation_sd: float = 0.1
    activation_selection: List[str] = Field(default_factory=lambda: ["ReLU", "ELU", "GELU"])
    mutate_elite: bool = True
    rand_seed: Optional[int] = None
    # per-hyperparameter mutation bounds: {hp_name: {min: .., max: ..}}
    # (reference manifest.py mutation.rl_hp_selection -> RLParameter ranges)
    rl_hp_selection: Dict[str, Dict[str, float]] = Field(default_factory=dict)


class TournamentSelectionSpec(_Base):
    tournament_size: int = 2
    elitism: bool = True


class ReplayBufferSpec(_Base):
    max_size: int = 100_000
    per: bool = False
    alpha: float = 0.6
    beta: float = 0.4
    n_step: int = 1
    storage_device: Optional[str] = None


class TrainingSpec(_Base):
    max_steps: int = 100_000
    pop_size: int = 4
    evo_steps: int = 10_000
    eval_steps: Optional[int] = None
    eval_loop: int = 1
    learning_delay: int = 0
    eps_start: float = 1.0
    eps_end: float = 0.05
    eps_decay: float = 0.995
    target: Optional[float] = None

    @model_validator(mode="before")
    @classmethod
    def _reference_aliases(cls, data):
        if isinstance(data, dict) and "target_score" in data and "target" not in data:
            data = dict(data)
            data["target"] = data.pop("target_score")
        return data
    checkpoint: Optional[int] = None
    checkpoint_path: Optional[str] = None
    overwrite_checkpoints: bool = True
    save_elite: bool = False
    elite_path: Optional[str] = None
    max_wall_seconds: Optional[float] = None
    resume_from_checkpoint: Optional[str] = None
    seed: Optional[int] = None


class TrainingManifest(_Base):
    """Top-level manifest document."""

    algorithm: AlgorithmSpec
    environment: Dict[str, Any] = Field(default_factory=dict)
    network: NetworkSpec = Field(default_factory=NetworkSpec)
    mutation: MutationSpec = Field(default_factory=MutationSpec)
    replay_buffer: ReplayBufferSpec = Field(default_factory=ReplayBufferSpec)
    selection_strategy: TournamentSelectionSpec = Field(default_factory=TournamentSelectionSpec)
    training: TrainingSpec = Field(default_factory=TrainingSpec)

    @model_validator(mode="after")
    def _check_algo(self):
        algo_workload(self.algorithm.name)
        return self

    def env_spec(self) -> EnvSpec:
        env = dict(self.environment)
        # reference manifests key the env as `name:` (configs/training/*)
        if "env_id" not in env and "name" in env:
            env["env_id"] = env.pop("name")
        etype = env.get("type")
        if etype is None:
            workload = algo_workload(self.algorithm.name)
            etype = {
                "off_policy": "gym",
                "on_policy": "gym",
                "offline": "offline",
                "multi_agent_off_policy": "pettingzoo",
                "multi_agent_on_policy": "pettingzoo",
                "bandit": "bandit",
            }.get(workload, "llm" if workload.startswith("llm") else "gym")
            env["type"] = etype
        cls = {
            "gym": GymEnvSpec,
            "pettingzoo": PzEnvSpec,
            "llm": LLMEnvSpec,
            "bandit": BanditEnvSpec,
            "offline": OfflineEnvSpec,
        }[etype]
        return cls.model_validate(env)

    @classmethod
    def from_yaml(cls, path: str) -> "TrainingManifest":
        import yaml

        with open(path) as f:
            return cls.model_validate(yaml.safe_load(f))

    def to_yaml(self, path: Optional[str] = None) -> str:
        import yaml

        text = yaml.safe_dump(self.model_dump(exclude_none=True), sort_keys=False)
        if path:
            with open(path, "w") as f:
                f.write(text)
        return text
