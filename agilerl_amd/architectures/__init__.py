from .llama_patches import apply_hip_kernels_to_llama

__all__ = ["apply_hip_kernels_to_llama"]
