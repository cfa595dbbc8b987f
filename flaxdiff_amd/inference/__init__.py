from .pipeline import DiffusionInferencePipeline, InferencePipeline
from .utils import (canonicalize_architecture, load_from_checkpoint,
                    map_nested_config, parse_config)

__all__ = ["DiffusionInferencePipeline", "InferencePipeline", "canonicalize_architecture",
           "load_from_checkpoint", "map_nested_config", "parse_config"]
