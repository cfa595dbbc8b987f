from .optim import FlatAdamWEMA, warmup_cosine_schedule
from .simple_trainer import SimpleTrainer
from .diffusion_trainer import DiffusionTrainer, l2_loss
from .general_diffusion_trainer import GeneralDiffusionTrainer
from .autoencoder_trainer import AutoEncoderTrainer

__all__ = ["FlatAdamWEMA", "warmup_cosine_schedule", "SimpleTrainer",
           "DiffusionTrainer", "GeneralDiffusionTrainer", "AutoEncoderTrainer",
           "l2_loss"]
