from .optim import FlatAdamWEMA, warmup_cosine_schedule
from .simple_trainer import SimpleTrainer
from .diffusion_trainer import DiffusionTrainer, l2_loss
from .general_diffusion_trainer import GeneralDiffusionTrainer, generate_modelname
from .autoencoder_trainer import AutoEncoderTrainer

__all__ = ["FlatAdamWEMA", "warmup_cosine_schedule", "SimpleTrainer",
           "DiffusionTrainer", "GeneralDiffusionTrainer", "generate_modelname",
           "AutoEncoderTrainer",
           "l2_loss"]
