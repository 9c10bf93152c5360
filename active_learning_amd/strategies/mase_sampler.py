"""MASE: margin in feature space (distance to linear decision boundaries).

Reference: src/query_strategies/mase_sampler.py. The reference builds
weight_delta (B,C,M) and epsilon (B,C,M) per batch (:59-79); here the
closed form |logit_p - logit_c| / ||w_p - w_c|| (ops/scoring.mase_margins)
produces identical margins from logits + the C x C weight Gram — no (B,C,M)
tensor. The boundary sanity check (:85-90) is preserved: perturbing the last
batch's embeddings by the optimal epsilon must land on the decision boundary
(top-2 logits equal within 1e-4).
"""

import torch

from ..ops.scoring import mase_margins
from .common import core_net, forward_pool
from .strategy import Strategy


class MASESampler(Strategy):
    def query(self, budget):
        idxs_for_query = self.available_query_idxs(boolean=False, shuffle=False)
        min_margins, _, _, _ = self.compute_margins(idxs_for_query)
        budget = int(min(len(idxs_for_query), budget))
        order = torch.sort(min_margins, descending=False).indices[:budget].cpu()
        labeled_idxs = idxs_for_query[order.numpy()].tolist()
        return labeled_idxs, budget

    @torch.no_grad()
    def compute_margins(self, idxs_for_query, use_training_augmentation=False):
        logits, embedding, labels = forward_pool(
            self, idxs_for_query, want_embedding=True,
            use_al_set=not use_training_augmentation)
        net = core_net(self.net)
        weight = net.linear.weight.detach().to(logits.device)
        min_margins, radius, pred = mase_margins(logits, weight)
        self._boundary_sanity_check(logits, embedding, radius, pred, weight, net)
        return (min_margins.cpu(), radius.cpu(), pred.cpu(), labels.cpu())

    def _boundary_sanity_check(self, logits, embedding, radius, pred, weight, net,
                               max_check=256):
        """Move each embedding by the optimal epsilon onto the nearest
        boundary; top-2 logits must then be equal (mase_sampler.py:85-90).
        epsilon = -(w_p - w_c*) * lam / 2, lam = 2 (l_p - l_c*) / ||dw||^2.
        """
        sl = slice(max(0, logits.shape[0] - max_check), logits.shape[0])
        lg, em, pr = logits[sl], embedding[sl], pred[sl]
        rad = radius[sl]
        cstar = rad.min(dim=1).indices
        b = torch.arange(lg.shape[0], device=lg.device)
        dw = weight[pr] - weight[cstar]  # (b, M)
        denom = (dw * dw).sum(dim=1).clamp_min(1e-30)
        lam = 2 * (lg[b, pr] - lg[b, cstar]) / denom
        eps = -dw * (lam / 2)[:, None]
        new_logits = net(em + eps, specify_input_layer="finalembed").float()
        top2 = torch.topk(new_logits, k=2, dim=1, largest=True).values
        gap = (top2[:, 0] - top2[:, 1]).abs().mean()
        assert gap < 1e-3, f"MASE boundary sanity check failed: mean top-2 gap {gap}"
