"""Margin sampling (smallest top1-top2 softmax gap).

Reference: src/query_strategies/margin_sampler.py:19-45; the softmax + top-2
margin is one fused kernel pass on GPU (ops/scoring.py).
"""

import torch

from ..ops.scoring import margin_scores
from .common import forward_pool
from .strategy import Strategy


class MarginSampler(Strategy):
    def query(self, budget):
        idxs_for_query = self.available_query_idxs(boolean=False, shuffle=False)
        logits, _, _ = forward_pool(self, idxs_for_query)
        margins = margin_scores(logits)
        budget = int(min(len(idxs_for_query), budget))
        order = torch.sort(margins, descending=False).indices[:budget].cpu()
        labeled_idxs = idxs_for_query[order.numpy()].tolist()
        self.net.train()
        return labeled_idxs, budget
