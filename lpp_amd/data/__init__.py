from .synthetic import SyntheticCausalLMDataset, CausalLMCollator, RepeatingLoader, build_loader

__all__ = [
    "SyntheticCausalLMDataset",
    "CausalLMCollator",
    "RepeatingLoader",
    "build_loader",
]
