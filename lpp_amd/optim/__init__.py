from .adamw import MixedPrecisionAdamW
from .scheduler import WarmupDecayLR

__all__ = ["MixedPrecisionAdamW", "WarmupDecayLR"]
