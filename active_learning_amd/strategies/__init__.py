"""Strategy factory — explicit registry (reference uses eval-based dispatch,
src/query_strategies/get_strategy.py:16-17)."""

from .badge_sampler import BADGESampler
from .balanced_random_sampler import BalancedRandomSampler
from .balancing_sampler import BalancingSampler
from .base_sampler import BASESampler
from .confidence_sampler import ConfidenceSampler
from .coreset_sampler import CoresetSampler
from .entropy_sampler import EntropySampler
from .margin_clustering_sampler import MarginClusteringSampler
from .margin_sampler import MarginSampler
from .mase_sampler import MASESampler
from .partitioned_badge_sampler import PartitionedBADGESampler
from .partitioned_coreset_sampler import PartitionedCoresetSampler
from .random_sampler import RandomSampler
from .strategy import Strategy
from .vaal_sampler import VAALSampler

STRATEGIES = {
    "RandomSampler": RandomSampler,
    "BalancedRandomSampler": BalancedRandomSampler,
    "ConfidenceSampler": ConfidenceSampler,
    "MarginSampler": MarginSampler,
    "EntropySampler": EntropySampler,
    "MASESampler": MASESampler,
    "BASESampler": BASESampler,
    "CoresetSampler": CoresetSampler,
    "BADGESampler": BADGESampler,
    "PartitionedCoresetSampler": PartitionedCoresetSampler,
    "PartitionedBADGESampler": PartitionedBADGESampler,
    "BalancingSampler": BalancingSampler,
    "MarginClusteringSampler": MarginClusteringSampler,
    "VAALSampler": VAALSampler,
}


def get_strategy(name: str):
    if name not in STRATEGIES:
        raise ValueError(f"Unknown strategy {name!r}; available: {sorted(STRATEGIES)}")
    return STRATEGIES[name]
