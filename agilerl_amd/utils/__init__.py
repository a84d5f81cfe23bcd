from .utils import (
    make_vect_envs,
    create_population,
    save_population_checkpoint,
    save_llm_checkpoint,
    run_selection_and_mutation,
    get_env_defined_actions,
    observation_space_channels_to_first,
    consolidate_mutations,
    log_gpu_memory_snapshot,
)

__all__ = [
    "make_vect_envs",
    "create_population",
    "save_population_checkpoint",
    "save_llm_checkpoint",
    "run_selection_and_mutation",
    "get_env_defined_actions",
    "observation_space_channels_to_first",
    "consolidate_mutations",
    "log_gpu_memory_snapshot",
]
