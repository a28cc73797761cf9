from .llama import (
    DecoderLayerPipe,
    KVCache,
    EmbeddingPipe,
    LMHeadPipe,
    LlamaForCausalLM,
    NormPipe,
    RMSNorm,
    deterministic_layer_init,
    get_layers_from_config,
    init_monolithic_weights,
    init_pipeline_weights,
    init_weights,
    loss_fn,
)

__all__ = [
    "DecoderLayerPipe",
    "KVCache",
    "EmbeddingPipe",
    "LMHeadPipe",
    "LlamaForCausalLM",
    "NormPipe",
    "RMSNorm",
    "get_layers_from_config",
    "init_weights",
    "loss_fn",
]
