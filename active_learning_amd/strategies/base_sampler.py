"""BASE: class-balanced MASE (reference: src/query_strategies/base_sampler.py).

Semantics preserved: the budget splits into per-class quotas of
floor(budget/C) + 1 for the first budget % C classes (:23-24); a sample's
score for class c is its min-margin when the model predicts c and its
distance to c's decision boundary otherwise (:28-29); classes claim their
quota in index order, each claiming the lowest-scoring samples not already
claimed (:31-32).

Implemented as a single (N, C) score matrix built in one scatter (the
predicted-class column of each row is overwritten with the row's min-margin)
plus a per-class masked top-k — no per-class torch.where re-materialization,
everything stays on the scoring device.
"""

import torch

from .mase_sampler import MASESampler


class BASESampler(MASESampler):
    def query(self, budget):
        pool = self.available_query_idxs(boolean=False, shuffle=False)
        min_margins, per_class_margins, pred_labels, _ = self.compute_margins(pool)
        budget = int(min(len(pool), budget))
        n = len(pool)

        # score matrix: boundary distances, with row i's pred-class entry
        # replaced by its min-margin
        scores = per_class_margins.clone()
        scores[torch.arange(n, device=scores.device), pred_labels] = min_margins

        quota = torch.full((self.num_classes,), budget // self.num_classes,
                           dtype=torch.long)
        quota[:budget % self.num_classes] += 1

        claimed = torch.zeros(n, dtype=torch.bool, device=scores.device)
        picks = []
        inf = torch.tensor(float("inf"), device=scores.device)
        for c in torch.nonzero(quota).flatten().tolist():
            col = torch.where(claimed, inf, scores[:, c])
            best = torch.topk(col, int(quota[c]), largest=False).indices
            claimed[best] = True
            picks.append(best)
        picks = torch.cat(picks)
        assert picks.numel() == int(claimed.sum()), "duplicate claim in BASE"
        return pool[picks.cpu().numpy()].tolist(), budget
