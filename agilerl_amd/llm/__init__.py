from .packing import pack_padded_batch, unpack_values
from .scheduler import WarmupCosineLR, create_warmup_cosine_scheduler
from .offload import activation_offload
from .chat import apply_chat_template
from .paged_cache import PagedKVCache
from .decode_engine import DecodeEngine
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
    "pack_padded_batch",
    "unpack_values",
    "WarmupCosineLR",
    "create_warmup_cosine_scheduler",
    "activation_offload",
    "apply_chat_template",
    "PagedKVCache",
    "DecodeEngine",
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
