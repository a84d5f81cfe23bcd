from .trainer import Trainer, LocalTrainer, ArenaTrainer
from .train_off_policy import train_off_policy, save_population_checkpoint
from .train_on_policy import train_on_policy
from .train_multi_agent_off_policy import train_multi_agent_off_policy
from .train_multi_agent_on_policy import train_multi_agent_on_policy
from .train_bandits import train_bandits
from .train_offline import train_offline, load_transitions_into_buffer, save_transitions, load_transitions, collect_transitions
from .train_distributed import train_on_policy_distributed

__all__ = [
    "Trainer",
    "LocalTrainer",
    "ArenaTrainer",
    "train_off_policy",
    "train_on_policy",
    "train_multi_agent_off_policy",
    "train_multi_agent_on_policy",
    "train_bandits",
    "train_offline",
    "load_transitions_into_buffer",
    "train_on_policy_distributed",
    "save_population_checkpoint",
]
