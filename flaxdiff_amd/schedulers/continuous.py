"""Continuous scheduler base (reference: schedulers/continuous.py:7-13)."""
from .common import NoiseScheduler


class ContinuousNoiseScheduler(NoiseScheduler):
    """General continuous noise scheduler: timesteps live in [0, 1]."""

    def __init__(self, *args, **kwargs):
        kwargs.pop("timesteps", None)
        super().__init__(1, *args, **kwargs)
