"""Entropy sampling (largest predictive entropy). Not in the reference's
sampler set but named by BASELINE.json config 2 ("entropy + margin sampling");
same fused softmax-scores kernel as margin/confidence."""

import torch

from ..ops.scoring import entropy_scores
from .common import forward_pool
from .strategy import Strategy


class EntropySampler(Strategy):
    def query(self, budget):
        idxs_for_query = self.available_query_idxs(boolean=False, shuffle=False)
        logits, _, _ = forward_pool(self, idxs_for_query)
        ent = entropy_scores(logits)
        budget = int(min(len(idxs_for_query), budget))
        order = torch.sort(ent, descending=True).indices[:budget].cpu()
        labeled_idxs = idxs_for_query[order.numpy()].tolist()
        self.net.train()
        return labeled_idxs, budget
