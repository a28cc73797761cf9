from .synthetic import SyntheticCausalLMDataset, CausalLMCollator, RepeatingLoader, build_loader
from .text import (
    IGNORE_INDEX,
    FlattenedGroupDataset,
    PlaceholderDataset,
    PromptResponseDataset,
    RoundRobinMixDataset,
    Seq2SeqToCausalLM,
    SimpleTokenizer,
    TextCollator,
    completion_labels,
    expand_special_tokenizer,
    load_prompt_response_data,
)

__all__ = [
    "SyntheticCausalLMDataset",
    "CausalLMCollator",
    "RepeatingLoader",
    "build_loader",
    "IGNORE_INDEX",
    "FlattenedGroupDataset",
    "PlaceholderDataset",
    "PromptResponseDataset",
    "RoundRobinMixDataset",
    "Seq2SeqToCausalLM",
    "SimpleTokenizer",
    "TextCollator",
    "completion_labels",
    "expand_special_tokenizer",
    "load_prompt_response_data",
]
