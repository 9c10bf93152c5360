"""BADGE: k-means++ seeding over gradient embeddings.

Reference: src/query_strategies/badge_sampler.py. The gradient embedding
g_i = (softmax(l_i) - onehot(argmax l_i)) (x) e_i is NEVER materialized here
(the reference builds the B x C x M outer product per batch, :40 — 262M
floats/batch at ImageNet scale). Distances come from the factorized Gram
<g_i,g_j> = (a_i.a_j)(e_i.e_j) (ops/scoring.badge_pairwise_sqdist); the
POOLING_H x (POOLING_AREA/POOLING_H) adaptive-pool variant (:9-10,41-44)
factorizes the same way because pooling an outer product pools each factor.
Selection = randomized greedy k-center (k-means++ seeding, :72-73).
"""

import numpy as np

from ..ops.scoring import badge_pairwise_sqdist, badge_vectors
from .common import forward_pool
from .coreset_sampler import CoresetSampler

POOLING_H = 16
POOLING_AREA = 512


class BADGESampler(CoresetSampler):
    def get_badge_vectors(self, idxs, use_adaptive_pool=False):
        logits, emb, _ = forward_pool(self, idxs, want_embedding=True)
        pool = None
        if use_adaptive_pool:
            ph = min(POOLING_H, logits.shape[1])
            pw = int(float(POOLING_AREA) / ph)
            pool = (ph, pw)
        return badge_vectors(logits, emb, pool=pool)

    def get_gradient_embeddings(self, idxs, use_adaptive_pool=False):
        """Materialized (B, C*M) gradient embeddings — kept for parity/tests;
        query() uses the factorized path."""
        a, e = self.get_badge_vectors(idxs, use_adaptive_pool)
        return (a[:, :, None] * e[:, None, :]).reshape(a.shape[0], -1)

    def query(self, budget):
        idxs_for_coreset = self.get_idxs_for_coreset()
        if self._use_cached_dist():
            pairwise = self.saved_pairwise_l2_dist
        else:
            a, e = self.get_badge_vectors(idxs_for_coreset)
            pairwise = badge_pairwise_sqdist(a, e)
        labeled_bool = self.already_labeled_idxs(boolean=True)[idxs_for_coreset]
        budget = int(min(self.available_query_idxs(boolean=True)[idxs_for_coreset].sum(),
                         budget))
        chosen = self.coreset(pairwise, labeled_bool, budget, randomize=True)
        labeled_idxs = np.asarray(idxs_for_coreset)[chosen].tolist()
        return labeled_idxs, len(labeled_idxs)
