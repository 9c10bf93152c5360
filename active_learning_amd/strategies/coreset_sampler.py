"""Coreset: greedy k-center over embedding distances.

Reference: src/query_strategies/coreset_sampler.py. MI355X-native changes:
the N x N fp32 distance matrix, the running min-distance vector and the
10k-iteration greedy loop all stay on device (the reference pulls embeddings
to CPU, :43-57, and runs the selection loop host-side, :77-104). Subset caps
(:21-41), distance-matrix caching under freeze_feature (:112-121) and
randomized (k-means++-style) selection semantics are preserved.
"""

import numpy as np
import torch

from ..ops.scoring import kcenter_greedy, pairwise_sqdist
from .common import forward_pool
from .strategy import Strategy


class CoresetSampler(Strategy):
    def __init__(self, train_set, al_set, net, train_args, eval_idxs, comet_experiment,
                 test_set=None, **kwargs):
        super().__init__(train_set, al_set, net, train_args, eval_idxs,
                         comet_experiment, test_set, **kwargs)
        self.saved_pairwise_l2_dist = None
        self.subset_labeled = kwargs.get("subset_labeled")
        self.subset_unlabeled = kwargs.get("subset_unlabeled")

    # -- subset selection (coreset_sampler.py:21-41) ----------------------- #
    def get_idxs_for_coreset(self, return_sep_idxs=False):
        idxs_for_query = self.available_query_idxs(boolean=False, shuffle=True)
        idxs_labeled = self.already_labeled_idxs(boolean=False, shuffle=True)

        if self.subset_labeled is not None:
            subset_labeled = min(self.subset_labeled, len(idxs_labeled))
            idxs_labeled = idxs_labeled[:subset_labeled]
        if self.subset_unlabeled is not None:
            if self.subset_labeled is not None:
                subset_unlabeled = (self.subset_labeled + self.subset_unlabeled
                                    - len(idxs_labeled))
            else:
                subset_unlabeled = self.subset_unlabeled
            subset_unlabeled = min(subset_unlabeled, len(idxs_for_query))
            idxs_for_query = idxs_for_query[:subset_unlabeled]

        idxs_for_coreset = sorted(idxs_for_query.tolist() + idxs_labeled.tolist())
        if return_sep_idxs:
            return idxs_for_coreset, idxs_labeled.tolist(), idxs_for_query.tolist()
        return idxs_for_coreset

    def get_embeddings(self, idxs):
        """Full-pool inference; embeddings stay in HBM (coreset_sampler.py:43-57
        pages them to CPU)."""
        _, emb, _ = forward_pool(self, idxs, want_embedding=True)
        return emb

    def get_pairwise_l2_dist(self, features):
        return pairwise_sqdist(features)

    def coreset(self, pairwise_l2_dist, labeled_indicator, query_count, randomize=False):
        labeled = torch.as_tensor(np.asarray(labeled_indicator, dtype=bool),
                                  device=pairwise_l2_dist.device)
        return kcenter_greedy(pairwise_l2_dist, labeled, int(query_count),
                              randomize=randomize)

    def _use_cached_dist(self):
        return (self.freeze_feature and self.saved_pairwise_l2_dist is not None
                and self.subset_unlabeled is None and self.subset_labeled is None)

    def query(self, budget):
        idxs_for_coreset = self.get_idxs_for_coreset()
        if self._use_cached_dist():
            pairwise = self.saved_pairwise_l2_dist
        else:
            embeddings = self.get_embeddings(idxs_for_coreset)
            pairwise = self.get_pairwise_l2_dist(embeddings)
            if self.freeze_feature:
                self.saved_pairwise_l2_dist = pairwise
        labeled_bool = self.already_labeled_idxs(boolean=True)[idxs_for_coreset]
        budget = int(min(self.available_query_idxs(boolean=True)[idxs_for_coreset].sum(),
                         budget))
        chosen = self.coreset(pairwise, labeled_bool, budget)
        labeled_idxs = np.asarray(idxs_for_coreset)[chosen].tolist()
        return labeled_idxs, len(labeled_idxs)

    def __getstate__(self):
        """Cached distance matrices may live on GPU; move to CPU for the
        experiment pickle (resume then reloads host-side)."""
        state = super().__getstate__()
        d = state.get("saved_pairwise_l2_dist")
        if torch.is_tensor(d) and d.is_cuda:
            state["saved_pairwise_l2_dist"] = d.cpu()
        return state
