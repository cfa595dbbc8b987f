"""EvaluationMetric dataclass (reference /root/reference/flaxdiff/metrics/common.py:5-18)."""
from dataclasses import dataclass
from typing import Callable


@dataclass
class EvaluationMetric:
    """function(generated [B,H,W,C] in [-1,1], batch) -> scalar; `name` becomes
    the `val/<name>` log key; higher_is_better drives best-tracking direction."""

    function: Callable
    name: str
    higher_is_better: bool = False
