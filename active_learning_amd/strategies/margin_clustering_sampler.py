"""Cluster-Margin: HAC clusters + round-robin min-margin picks.

Reference: src/query_strategies/margin_clustering_sampler.py — embeddings +
margins in one pool pass (:23-45), sklearn AgglomerativeClustering
n_clusters=20 on the unlabeled embeddings (:59, host-side; run once unless
subsetting), round-robin over clusters sorted smallest-first taking the
min-margin sample of each (:71-87); consumed samples marked -1 and the
assignment persists across rounds (:80,89).
"""

import numpy as np

from ..ops.scoring import margin_scores
from .common import forward_pool
from .strategy import Strategy


class MarginClusteringSampler(Strategy):
    def __init__(self, train_set, al_set, net, train_args, eval_idxs, comet_experiment,
                 test_set=None, **kwargs):
        super().__init__(train_set, al_set, net, train_args, eval_idxs,
                         comet_experiment, test_set, **kwargs)
        self.subset_unlabeled = kwargs.get("subset_unlabeled")
        self.cluster_assignment = None
        self.n_clusters = 20

    def get_embeddings_and_margins(self, idxs):
        logits, emb, _ = forward_pool(self, idxs, want_embedding=True)
        margins = margin_scores(logits)
        return emb.cpu(), margins.cpu()

    def query(self, budget):
        if self.subset_unlabeled is None:
            idxs_for_hac = self.available_query_idxs(boolean=False, shuffle=False)
        else:
            idxs_for_hac = np.array(sorted(
                self.available_query_idxs(boolean=False,
                                          shuffle=True)[:self.subset_unlabeled]))
        embeddings, output_margins = self.get_embeddings_and_margins(idxs_for_hac)
        if self.cluster_assignment is None or self.subset_unlabeled:
            from sklearn.cluster import AgglomerativeClustering
            n_clusters = min(self.n_clusters, len(idxs_for_hac))
            cluster_assignment = AgglomerativeClustering(
                n_clusters=n_clusters).fit(embeddings.numpy()).labels_
        else:
            cluster_assignment = self.cluster_assignment

        cluster_ids, cluster_count = np.unique(cluster_assignment, return_counts=True)
        cluster_ids_sorted = [cid for _, cid in sorted(zip(cluster_count, cluster_ids))]
        margins = output_margins.numpy()

        query_idxs = []
        query_count = 0
        start_cluster = 0
        budget = int(min(len(idxs_for_hac), budget))
        while query_count < budget:
            for i in range(start_cluster, len(cluster_ids_sorted)):
                cid = cluster_ids_sorted[i]
                members = np.where(cluster_assignment == cid)[0]
                if len(members) == 0:
                    start_cluster += 1
                    continue
                pick = members[np.argmin(margins[members])]
                cluster_assignment[pick] = -1
                query_idxs.append(int(idxs_for_hac[pick]))
                query_count += 1
                if len(members) == 1:
                    start_cluster += 1
                if query_count >= budget:
                    break

        self.cluster_assignment = cluster_assignment[cluster_assignment != -1]
        return query_idxs, budget
