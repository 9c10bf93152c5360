"""Least-confidence sampling (smallest top softmax probability).

Reference: src/query_strategies/confidence_sampler.py:18-47. The reference
has a known indexing bug at :41 (re-indexes the per-subset confidence vector
with pool-level indices — wrong/out-of-range after round 0, SURVEY.md §2.2
row 18). This implements the INTENDED behavior (identical to MarginSampler's
correct pattern) and documents the divergence.
"""

import torch

from ..ops.scoring import confidence_scores
from .common import forward_pool
from .strategy import Strategy


class ConfidenceSampler(Strategy):
    def query(self, budget):
        idxs_for_query = self.available_query_idxs()
        logits, _, _ = forward_pool(self, idxs_for_query)
        confidence = confidence_scores(logits)
        query_count = int(min(len(idxs_for_query), budget))
        order = torch.sort(confidence, descending=False).indices[:query_count].cpu()
        labeled_idxs = idxs_for_query[order.numpy()].tolist()
        self.net.train()
        return labeled_idxs, query_count
