"""lpp_amd — MI355X-native LLaMA pipeline-parallel training engine.

A from-scratch framework with the capabilities of
SparkJiao/llama-pipeline-parallel, built MI355X-first:
PyTorch-ROCm + hand-written HIP/CDNA4 (gfx950) kernels for the hot path,
RCCL over xGMI for inter-stage p2p and data-parallel collectives, one
process per GPU, no DeepSpeed, no CUDA-compat shims.
"""

__version__ = "0.1.0"

from .config import ModelConfig, OptimizerConfig, TrainConfig, model_config
from .topology import ProcessGrid
from .layer_spec import LayerSpec, partition_balanced, partition_uniform
from .pipeline_module import PipelineModule
from .engine import PipelineEngine

__all__ = [
    "ModelConfig",
    "OptimizerConfig",
    "TrainConfig",
    "model_config",
    "ProcessGrid",
    "LayerSpec",
    "partition_balanced",
    "partition_uniform",
    "PipelineModule",
    "PipelineEngine",
]
