from .base import (
    EvolvableModule,
    EvolvableWrapper,
    DummyEvolvable,
    ModuleDict,
    MutationType,
    mutation,
    preserve_parameters,
    module_checkpoint_dict,
    load_module_from_checkpoint,
)
from .components import (
    NoisyLinear,
    GumbelSoftmax,
    NewGELU,
    ResidualBlock,
    SimbaResidualBlock,
    get_activation,
)
from .mlp import EvolvableMLP, create_mlp
from .cnn import EvolvableCNN
from .lstm import EvolvableLSTM
from .simba import EvolvableSimBa
from .multi_input import EvolvableMultiInput
from .gpt import EvolvableGPT, CausalSelfAttention, GPTBlock
from .bert import EvolvableBERT
from .resnet import EvolvableResNet

__all__ = [
    "EvolvableModule",
    "EvolvableWrapper",
    "DummyEvolvable",
    "ModuleDict",
    "MutationType",
    "mutation",
    "preserve_parameters",
    "module_checkpoint_dict",
    "load_module_from_checkpoint",
    "NoisyLinear",
    "GumbelSoftmax",
    "NewGELU",
    "ResidualBlock",
    "SimbaResidualBlock",
    "get_activation",
    "EvolvableMLP",
    "create_mlp",
    "EvolvableCNN",
    "EvolvableLSTM",
    "EvolvableSimBa",
    "EvolvableMultiInput",
    "EvolvableGPT",
    "CausalSelfAttention",
    "GPTBlock",
    "EvolvableBERT",
    "EvolvableResNet",
]
