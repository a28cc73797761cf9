"""Microbatch/memory schedule policy for MI355X (288 GB HBM3E).

Chooses (micro_batch_size, gradient_accumulation_steps,
checkpoint_layers_per_stage) per pipeline depth:

- larger microbatches raise GEMM/kernel efficiency (~+8% at mbs 4) but
  multiply in-flight activation memory by the pipeline depth and shrink
  the microbatch count (pipeline bubble (P-1)/(M+P-1)) at fixed tokens;
- activation recompute costs ~+33% forward work per checkpointed layer,
  so checkpoint only as many layers per stage as the byte budget demands
  (usually zero — the reference's always-on checkpointing is an
  80 GB-HBM coping mechanism, conf/...yaml:19).

The byte estimator is validated against measured peaks (±1 GB at the 65B
shapes, profiles/README.md).
"""

from __future__ import annotations

from dataclasses import dataclass

# Leave ~24 GB of the 288 GB HBM3E for logits/p2p/allocator slack.  The
# byte estimator itself is ~10 GB conservative vs measured peaks (r01/r02
# bench peak_mem_gb), so the real slack is ~34 GB at the 65B shapes.
HBM_BUDGET_BYTES = 264e9
BYTES_PER_PARAM = 20      # bf16 param + fp32 master/exp_avg/exp_avg_sq/grad
                          # + bf16 W^T dgrad copy (ops/linear._weight_t)


@dataclass
class Schedule:
    micro_batch_size: int
    gas: int
    ckpt_layers_per_stage: int


def act_bytes_per_layer(seq_len: int, hidden: int, intermediate: int,
                        micro_batch_size: int) -> int:
    """Saved activations of one NON-checkpointed decoder layer (bf16):
    ~8 S*H tensors (norm inputs, q/k/v, attention out, lse) + 3 S*I
    (gate/up/swiglu-out)."""
    return (8 * seq_len * hidden + 3 * seq_len * intermediate) * micro_batch_size * 2


def choose_schedule(model_cfg, num_stages: int, layers_per_stage: int,
                    seq_len: int, micro_batch_size: int = 0, gas: int = 0,
                    budget: float = HBM_BUDGET_BYTES) -> Schedule:
    """mbs/gas of 0 mean "auto"."""
    if micro_batch_size == 0:
        micro_batch_size = {1: 4, 2: 4, 4: 2}.get(num_stages, 1)
    if gas == 0:
        gas = max(num_stages * 16, 64 // micro_batch_size)
    apl = act_bytes_per_layer(seq_len, model_cfg.hidden_size,
                              model_cfg.intermediate_size, micro_batch_size)
    in_flight = min(num_stages, gas)  # stage 0 holds the most microbatches
    # Worst-stage parameter bytes: an even slice, plus the embedding OR the
    # LM head for the edge stages when partitioned (num_params() already
    # counts both, so a single-stage run adds nothing — double-counting
    # them forced needless recompute at 128k-vocab models).
    vocab_extra = (model_cfg.vocab_size * model_cfg.hidden_size
                   if num_stages > 1 else 0)
    stage_params = model_cfg.num_params() // max(num_stages, 1) + vocab_extra
    free_layers = max(0, int((budget - stage_params * BYTES_PER_PARAM)
                             // (apl * in_flight)))
    ckpt = max(0, layers_per_stage - free_layers)
    return Schedule(micro_batch_size, gas, ckpt)
