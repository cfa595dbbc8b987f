from .common import DiffusionSampler
from .ddpm import DDPMSampler, SimpleDDPMSampler
from .ddim import DDIMSampler
from .euler import EulerSampler, SimplifiedEulerSampler, EulerAncestralSampler
from .heun_sampler import HeunSampler
from .rk4_sampler import RK4Sampler
from .multistep_dpm import MultiStepDPM

__all__ = [
    "DiffusionSampler", "DDPMSampler", "SimpleDDPMSampler", "DDIMSampler",
    "EulerSampler", "SimplifiedEulerSampler", "EulerAncestralSampler",
    "HeunSampler", "RK4Sampler", "MultiStepDPM",
]
