from .base import (EvolvableNetwork, build_encoder, get_default_encoder_config,
                   preprocess_observation, CustomNetworkAdapter)
from .distributions import ActionDistribution, EvolvableDistribution
from .q_networks import QNetwork, RainbowQNetwork, ContinuousQNetwork
from .actors import DeterministicActor, StochasticActor
from .value_networks import ValueNetwork

__all__ = [
    "EvolvableNetwork",
    "build_encoder",
    "get_default_encoder_config",
    "preprocess_observation",
    "CustomNetworkAdapter",
    "ActionDistribution",
    "EvolvableDistribution",
    "QNetwork",
    "RainbowQNetwork",
    "ContinuousQNetwork",
    "DeterministicActor",
    "StochasticActor",
    "ValueNetwork",
]
