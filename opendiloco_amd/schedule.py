"""Learning-rate schedule.

Re-implements ``transformers.get_cosine_schedule_with_warmup`` exactly
(the reference uses it at open_diloco/train_fsdp.py:255-260 and
train_diloco_torch.py:189-193).  lr traces must be bit-exact with the
reference's (tests compare lr with ``==``,
reference tests/test_training/test_train.py:83,206).
"""

from __future__ import annotations

import math

from torch.optim.lr_scheduler import LambdaLR


def _cosine_with_warmup_lambda(current_step: int, *, num_warmup_steps: int, num_training_steps: int,
                               num_cycles: float = 0.5) -> float:
    # identical formula to transformers.optimization._get_cosine_schedule_with_warmup_lr_lambda
    if current_step < num_warmup_steps:
        return float(current_step) / float(max(1, num_warmup_steps))
    progress = float(current_step - num_warmup_steps) / float(max(1, num_training_steps - num_warmup_steps))
    return max(0.0, 0.5 * (1.0 + math.cos(math.pi * float(num_cycles) * 2.0 * progress)))


def get_cosine_schedule_with_warmup(optimizer, num_warmup_steps: int, num_training_steps: int,
                                    num_cycles: float = 0.5, last_epoch: int = -1) -> LambdaLR:
    def lr_lambda(current_step: int) -> float:
        return _cosine_with_warmup_lambda(
            current_step,
            num_warmup_steps=num_warmup_steps,
            num_training_steps=num_training_steps,
            num_cycles=num_cycles,
        )

    return LambdaLR(optimizer, lr_lambda, last_epoch)
