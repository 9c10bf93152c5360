"""BASE: class-balanced MASE.

Reference: src/query_strategies/base_sampler.py — per-class budget
budget/C (+1 for the first budget%C classes, :23-24); per class c the score
is the min margin where predicted==c, else the distance to c's boundary
(:28-29); already-chosen masked with inf (:31-32); uniqueness asserted (:37).
"""

import torch

from .mase_sampler import MASESampler


class BASESampler(MASESampler):
    def query(self, budget):
        idxs_for_query = self.available_query_idxs(boolean=False, shuffle=False)
        min_margins, per_class_margins, pred_labels, _ = self.compute_margins(
            idxs_for_query)
        budget = int(min(len(idxs_for_query), budget))

        labeled_idxs = []
        for c in range(self.num_classes):
            take = budget // self.num_classes + int(c < budget % self.num_classes)
            if take == 0:
                continue
            dist_c = torch.where(pred_labels == c, min_margins, per_class_margins[:, c])
            if labeled_idxs:
                dist_c = dist_c.clone()
                dist_c[torch.tensor(labeled_idxs)] = float("inf")
            order = torch.sort(dist_c, descending=False).indices
            labeled_idxs += order[:take].tolist()

        assert len(labeled_idxs) == len(set(labeled_idxs))
        labeled_idxs = idxs_for_query[labeled_idxs].tolist()
        return labeled_idxs, budget
