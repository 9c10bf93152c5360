"""Partitioned BADGE: partitioned randomized k-center over pooled gradient
embeddings.

Reference: src/query_strategies/partitioned_badge_sampler.py (diamond
inheritance (BADGESampler, PartitionedCoresetSampler), pooled gradient
embeddings + randomized coreset per partition). The per-partition distance
matrix uses the factorized BADGE Gram, never the (B, 512) materialized
embedding.
"""

from ..ops.scoring import badge_pairwise_sqdist
from .badge_sampler import BADGESampler
from .partitioned_coreset_sampler import PartitionedCoresetSampler


class PartitionedBADGESampler(BADGESampler, PartitionedCoresetSampler):
    def query(self, budget):
        return self._query_with_embedding_func(budget, None, randomize_coreset=True)

    def _partition_pairwise(self, _embed_f, part_idxs):
        a, e = self.get_badge_vectors(list(part_idxs), use_adaptive_pool=True)
        return badge_pairwise_sqdist(a, e)
