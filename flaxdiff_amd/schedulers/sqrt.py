"""Sqrt continuous schedule (reference: schedulers/sqrt.py:7-11)."""
import torch

from .common import reshape_rates
from .continuous import ContinuousNoiseScheduler


class SqrtContinuousNoiseScheduler(ContinuousNoiseScheduler):
    def get_rates(self, steps, shape=(-1, 1, 1, 1)):
        if not torch.is_tensor(steps):
            steps = torch.as_tensor(steps, dtype=torch.float32)
        steps = steps.float()
        return reshape_rates((torch.sqrt(1 - steps), torch.sqrt(steps)), shape=shape)
