from .common import (
    NoiseScheduler,
    GeneralizedNoiseScheduler,
    get_coeff_shapes_tuple,
    reshape_rates,
)
from .discrete import DiscreteNoiseScheduler
from .linear import LinearNoiseSchedule, linear_beta_schedule
from .cosine import (
    CosineNoiseScheduler,
    CosineGeneralNoiseScheduler,
    CosineContinuousNoiseScheduler,
    cosine_beta_schedule,
)
from .exp import ExpNoiseSchedule, exp_beta_schedule
from .sqrt import SqrtContinuousNoiseScheduler
from .continuous import ContinuousNoiseScheduler
from .karras import (
    KarrasVENoiseScheduler,
    EDMNoiseScheduler,
    SimpleExpNoiseScheduler,
)

__all__ = [
    "NoiseScheduler",
    "GeneralizedNoiseScheduler",
    "DiscreteNoiseScheduler",
    "LinearNoiseSchedule",
    "CosineNoiseScheduler",
    "CosineGeneralNoiseScheduler",
    "CosineContinuousNoiseScheduler",
    "ExpNoiseSchedule",
    "SqrtContinuousNoiseScheduler",
    "ContinuousNoiseScheduler",
    "KarrasVENoiseScheduler",
    "EDMNoiseScheduler",
    "SimpleExpNoiseScheduler",
    "get_coeff_shapes_tuple",
    "reshape_rates",
    "linear_beta_schedule",
    "cosine_beta_schedule",
    "exp_beta_schedule",
]
