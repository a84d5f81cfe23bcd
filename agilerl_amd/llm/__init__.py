from .lora import (
    LoraConfig,
    LoraLinear,
    apply_lora,
    set_active_adapter,
    add_adapter,
    adapter_state_dict,
    load_adapter_state_dict,
    save_adapter,
    load_adapter,
    mark_only_adapter_trainable,
)

__all__ = [
    "LoraConfig",
    "LoraLinear",
    "apply_lora",
    "set_active_adapter",
    "add_adapter",
    "adapter_state_dict",
    "load_adapter_state_dict",
    "save_adapter",
    "load_adapter",
    "mark_only_adapter_trainable",
]
