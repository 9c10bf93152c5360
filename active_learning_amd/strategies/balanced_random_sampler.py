"""Class-balanced random baseline (cheats by reading labels).

Reference: src/query_strategies/balanced_random_sampler.py — per-class
budgets as equal as availability allows (threshold-raising there, :50-72;
shared water-filling allocator here), then a seeded shuffle pick per class.
"""

import numpy as np

from ..utils.pool_init import _balanced_allocation
from .strategy import Strategy


class BalancedRandomSampler(Strategy):
    """ONLY a baseline: peeks at labels of unqueried samples (reference
    docstring, balanced_random_sampler.py:8-11)."""

    def query(self, budget):
        labels = np.asarray(self.al_set.targets)
        avail = self.available_query_idxs(boolean=True)
        budget = int(min(avail.sum(), budget))

        counts = np.bincount(labels[avail], minlength=self.num_classes)
        alloc = _balanced_allocation(counts, budget)
        assert alloc.sum() == budget and (alloc <= counts).all()

        labeled_idxs = []
        for c in range(self.num_classes):
            if alloc[c] == 0:
                continue
            cand = np.where((labels == c) & avail)[0]
            cand = np.random.permutation(cand)[:alloc[c]]
            labeled_idxs += cand.tolist()
        assert np.unique(labeled_idxs).shape[0] == budget
        self.logger.info(f"Number of queried images: {budget}")
        return labeled_idxs, budget
