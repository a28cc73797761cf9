"""WarmupDecayLR: linear warmup -> linear decay to 0 over total_num_steps.

Same two-phase shape as the DeepSpeed scheduler the reference configures
(conf/...yaml:130-135, step counts patched in at trainer_base_ds_mp.py:273-275).
"""

from __future__ import annotations


class WarmupDecayLR:
    def __init__(self, optimizer, warmup_num_steps: int, total_num_steps: int,
                 warmup_max_lr: float, warmup_min_lr: float = 0.0):
        self.optimizer = optimizer
        self.warmup_num_steps = max(1, warmup_num_steps)
        self.total_num_steps = max(self.warmup_num_steps + 1, total_num_steps)
        self.warmup_max_lr = warmup_max_lr
        self.warmup_min_lr = warmup_min_lr
        self.last_step = 0
        self._apply()

    def get_lr(self) -> float:
        t = self.last_step
        if t < self.warmup_num_steps:
            frac = t / self.warmup_num_steps
            return self.warmup_min_lr + (self.warmup_max_lr - self.warmup_min_lr) * frac
        frac = (self.total_num_steps - t) / (self.total_num_steps - self.warmup_num_steps)
        return max(0.0, self.warmup_max_lr * frac)

    def _apply(self) -> None:
        self.optimizer.lr = self.get_lr()

    def step(self) -> None:
        self.last_step += 1
        self._apply()

    def state_dict(self) -> dict:
        return {"last_step": self.last_step}

    def load_state_dict(self, sd: dict) -> None:
        self.last_step = sd["last_step"]
        self._apply()
