"""Per-algorithm default hyperparameters.

Reference parity: ``agilerl/models/algorithms/`` (one spec file per
algorithm carrying defaults, e.g. ``dqn.py:30`` ``cudagraphs``).  Here a
single registry of default dicts that ``TrainingManifest`` merges under
user-provided hyperparameters.
"""

from __future__ import annotations

from typing import Any, Dict

__all__ = ["ALGO_DEFAULTS", "merged_hyperparameters"]

ALGO_DEFAULTS: Dict[str, Dict[str, Any]] = {
    "DQN": {"batch_size": 64, "lr": 1e-4, "gamma": 0.99, "tau": 1e-3,
            "double": False, "cudagraphs": False, "learn_step": 5},
    "RainbowDQN": {"batch_size": 64, "lr": 1e-4, "gamma": 0.99, "tau": 1e-3,
                   "n_step": 3, "num_atoms": 51, "v_min": -10.0, "v_max": 10.0,
                   "learn_step": 5},
    "CQN": {"batch_size": 64, "lr": 1e-4, "gamma": 0.99, "tau": 1e-3,
            "cql_alpha": 1.0, "double": True},
    "DDPG": {"batch_size": 64, "lr_actor": 1e-4, "lr_critic": 1e-3,
             "gamma": 0.99, "tau": 1e-3, "O_U_noise": True, "expl_noise": 0.1},
    "TD3": {"batch_size": 64, "lr_actor": 1e-4, "lr_critic": 1e-3,
            "gamma": 0.99, "tau": 5e-3, "policy_freq": 2, "policy_noise": 0.2},
    "PPO": {"batch_size": 512, "lr": 3e-4, "learn_step": 128, "gamma": 0.99,
            "gae_lambda": 0.95, "clip_coef": 0.2, "ent_coef": 0.01,
            "vf_coef": 0.5, "update_epochs": 4, "max_grad_norm": 0.5},
    "MADDPG": {"batch_size": 64, "lr_actor": 1e-4, "lr_critic": 1e-3,
               "gamma": 0.95, "tau": 1e-2},
    "MATD3": {"batch_size": 64, "lr_actor": 1e-4, "lr_critic": 1e-3,
              "gamma": 0.95, "tau": 1e-2, "policy_freq": 2},
    "IPPO": {"batch_size": 512, "lr": 3e-4, "learn_step": 128, "gamma": 0.99,
             "gae_lambda": 0.95, "clip_coef": 0.2, "ent_coef": 0.01},
    "NeuralUCB": {"batch_size": 64, "lr": 1e-3, "gamma": 1.0, "lamb": 1.0},
    "NeuralTS": {"batch_size": 64, "lr": 1e-3, "gamma": 1.0, "lamb": 1.0},
    "GRPO": {"group_size": 8, "lr": 5e-6, "clip_coef": 0.2, "beta": 0.04,
             "update_epochs": 1, "micro_batch_size": 2, "loss_norm": "token"},
    "GSPO": {"group_size": 8, "lr": 5e-6, "clip_coef": 0.2, "beta": 0.0,
             "update_epochs": 1, "micro_batch_size": 2},
    "CISPO": {"group_size": 8, "lr": 5e-6, "clip_coef": 0.2, "beta": 0.0,
              "update_epochs": 1, "micro_batch_size": 2},
    "PPOLLM": {"lr": 5e-6, "clip_coef": 0.2, "vf_coef": 0.5,
               "micro_batch_size": 2},
    "ReinforceLLM": {"group_size": 8, "lr": 5e-6, "micro_batch_size": 2},
    "SFT": {"lr": 1e-5, "micro_batch_size": 4},
    "DPO": {"lr": 5e-6, "beta": 0.1, "micro_batch_size": 2},
}


def merged_hyperparameters(name: str, user: Dict[str, Any]) -> Dict[str, Any]:
    out = dict(ALGO_DEFAULTS.get(name, {}))
    out.update(user)
    return out
