"""Balancing sampler (WACV'20 imbalanced AL).

Reference: src/query_strategies/balancing_sampler.py — one sample at a time
(:61-134): if the labeled-class histogram is imbalanced relative to the
remaining budget (:83-84), pick the unlabeled point minimizing
dist-to-rarest-centroid / max-dist-to-majority-centroids (:86-125); else
random (:128). Embedding cache under freeze_feature (:34-57). Embeddings and
centroid distances stay on device here.
"""

import numpy as np
import torch

from ..ops.scoring import class_centroids, sqdist_to_centers
from .common import forward_pool
from .strategy import Strategy


class BalancingSampler(Strategy):
    def __init__(self, train_set, al_set, net, train_args, eval_idxs, comet_experiment,
                 test_set=None, **kwargs):
        super().__init__(train_set, al_set, net, train_args, eval_idxs,
                         comet_experiment, test_set, **kwargs)
        self.saved_embeddings = None
        self.saved_ys = None

    def query(self, budget):
        self.feature_net = self.net
        idxs_for_query = self.available_query_idxs(boolean=True)
        idxs_labeled = self.already_labeled_idxs(boolean=True)
        labeled_idxs_cur_rd = []

        if self.freeze_feature and self.saved_embeddings is not None:
            embeddings, ys = self.saved_embeddings, self.saved_ys
        else:
            _, embeddings, ys = forward_pool(self, np.arange(self.n_pool),
                                             want_embedding=True)
            if self.freeze_feature:
                self.saved_embeddings, self.saved_ys = embeddings, ys
        device = embeddings.device
        ys = ys.to(device)

        budget = int(min(idxs_for_query.sum(), budget))
        q_mask = torch.as_tensor(idxs_for_query, device=device)
        l_mask = torch.as_tensor(idxs_labeled, device=device)

        for qi in range(budget):
            ys_labeled = ys[l_mask]
            counts = torch.bincount(ys_labeled, minlength=self.num_classes).float()
            mean_count = counts.mean()
            maj = counts > mean_count
            minor = ~maj
            maj_avg = counts[maj].sum() / maj.sum().clamp_min(1)
            minor_avg = counts[minor].sum() / minor.sum().clamp_min(1)

            remaining = budget - qi
            if remaining <= (minor.sum() * (maj_avg - minor_avg)).item():
                centers = class_centroids(embeddings[l_mask], ys_labeled,
                                          self.num_classes)
                rarest_count, rarest = counts.min(dim=0)
                emb_u = embeddings[q_mask]
                d_rare = sqdist_to_centers(emb_u, centers[rarest][None, :])
                if rarest_count == 0:
                    d_rare = torch.ones_like(d_rare)
                d_maj = sqdist_to_centers(emb_u, centers[maj])
                max_d_maj = d_maj.max(dim=1, keepdim=True).values
                score = (d_rare / max_d_maj).squeeze(1)
                local = int(score.min(dim=0).indices.item())
                query_idx = int(q_mask.nonzero(as_tuple=True)[0][local].item())
            else:
                avail = q_mask.nonzero(as_tuple=True)[0]
                query_idx = int(avail[torch.randint(len(avail), (1,))].item())

            q_mask[query_idx] = False
            l_mask[query_idx] = True
            labeled_idxs_cur_rd.append(query_idx)
        return labeled_idxs_cur_rd, len(labeled_idxs_cur_rd)

    def __getstate__(self):
        state = super().__getstate__()
        for k in ("saved_embeddings", "saved_ys"):
            v = state.get(k)
            if torch.is_tensor(v) and v.is_cuda:
                state[k] = v.cpu()
        return state
