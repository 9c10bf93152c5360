"""Shared query-pass helpers: batched inference over pool subsets with
device-resident collection (the reference pages every batch back to CPU,
e.g. coreset_sampler.py:43-57; on MI355X the pool caches stay in HBM)."""

import torch
from torch.utils.data import DataLoader, Subset


@torch.no_grad()
def forward_pool(strategy, idxs, want_embedding=False, use_al_set=True, keep_device=True):
    """Run strategy.net over Subset(al_set, idxs).

    Returns (logits, embeddings or None, labels); tensors stay on
    strategy.device when keep_device (fp32).
    """
    dataset = strategy.al_set if use_al_set else strategy.train_set
    loader = DataLoader(Subset(dataset, indices=list(idxs)), shuffle=False,
                        **strategy.train_args["loader_te_args"], drop_last=False)
    net = strategy.net
    net.eval()
    net.to(strategy.device)
    logits_l, emb_l, y_l = [], [], []
    for x, y, _ in loader:
        x = x.to(strategy.device, non_blocking=True)
        if want_embedding:
            out, emb = net(x, return_features="finalembed")
            emb_l.append(emb.float() if keep_device else emb.float().cpu())
        else:
            out = net(x)
        logits_l.append(out.float() if keep_device else out.float().cpu())
        y_l.append(y)
    logits = torch.cat(logits_l, dim=0)
    emb = torch.cat(emb_l, dim=0) if emb_l else None
    labels = torch.cat(y_l, dim=0)
    return logits, emb, labels


def core_net(net):
    """Unwrap a DDP wrapper (reference: hasattr(net,'module') check,
    mase_sampler.py:46-49)."""
    return net.module if hasattr(net, "module") else net
