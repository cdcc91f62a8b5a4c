from easyparallellibrary_amd.strategies.base import ParallelStrategy
from easyparallellibrary_amd.strategies.context import StrategyContext
from easyparallellibrary_amd.strategies.replicate import Replicate, replicate
from easyparallellibrary_amd.strategies.split import Split, split

__all__ = [
    "ParallelStrategy", "StrategyContext",
    "Replicate", "replicate", "Split", "split",
]
