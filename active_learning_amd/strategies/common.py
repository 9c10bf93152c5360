"""Shared query-pass helpers: batched inference over pool subsets with
device-resident collection (the reference pages every batch back to CPU,
e.g. coreset_sampler.py:43-57; on MI355X the pool caches stay in HBM).

Multi-GPU query sharding: the reference's query pass is single-GPU by design
(every sampler forwards the whole pool on device 0 while the other GPUs sit
idle between training rounds). Here forward_pool shards the pool across all
visible GPUs — one spawned worker per device, contiguous shards, results
gathered through files — whenever the strategy trains with world_size > 1
and the pool is large enough to amortize the spawn (AL_SHARD_QUERY_MIN,
default 4096 samples). Kill switch: AL_SHARD_QUERY=0.
"""

import copy
import os
import tempfile

import numpy as np
import torch
import torch.multiprocessing as mp
from torch.utils.data import DataLoader, Subset


@torch.no_grad()
def forward_pool(strategy, idxs, want_embedding=False, use_al_set=True, keep_device=True):
    """Run strategy.net over Subset(al_set, idxs).

    Returns (logits, embeddings or None, labels); tensors stay on
    strategy.device when keep_device (fp32).
    """
    dataset = strategy.al_set if use_al_set else strategy.train_set
    if _should_shard(strategy, idxs):
        return _sharded_forward_pool(strategy, dataset, idxs, want_embedding,
                                     keep_device)
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


def _should_shard(strategy, idxs):
    if os.environ.get("AL_SHARD_QUERY", "1") == "0":
        return False
    world = getattr(strategy, "world_size", 1) or 1
    if world < 2:
        return False
    if torch.cuda.is_available() and torch.cuda.device_count() < 2:
        return False
    min_n = int(os.environ.get("AL_SHARD_QUERY_MIN", "4096"))
    return len(idxs) >= min_n


def _shard_infer_worker(rank, world, net, dataset, shards, loader_args,
                        want_embedding, out_dir):
    """One process per device: forward a contiguous pool shard, save to disk."""
    if torch.cuda.is_available():
        device = torch.device("cuda", rank % torch.cuda.device_count())
    else:
        device = torch.device("cpu")
    net = net.to(device)
    net.eval()
    loader = DataLoader(Subset(dataset, indices=list(shards[rank])),
                        shuffle=False, drop_last=False, **loader_args)
    logits_l, emb_l, y_l = [], [], []
    with torch.no_grad():
        for x, y, _ in loader:
            x = x.to(device, non_blocking=True)
            if want_embedding:
                out, emb = net(x, return_features="finalembed")
                emb_l.append(emb.float().cpu())
            else:
                out = net(x)
            logits_l.append(out.float().cpu())
            y_l.append(y)
    torch.save({"logits": torch.cat(logits_l, dim=0),
                "emb": torch.cat(emb_l, dim=0) if emb_l else None,
                "labels": torch.cat(y_l, dim=0)},
               os.path.join(out_dir, f"shard_{rank}.pt"))


def _sharded_forward_pool(strategy, dataset, idxs, want_embedding, keep_device):
    """Fan the pool forward pass out over every visible GPU.

    Workers write CPU tensors to a temp dir and the parent concatenates in
    rank order (np.array_split preserves ordering), so the result is
    elementwise identical to the single-device path.
    """
    world = getattr(strategy, "world_size", 1) or 1
    if torch.cuda.is_available():
        world = min(world, torch.cuda.device_count())
    shards = [list(map(int, s)) for s in np.array_split(np.asarray(idxs), world)]
    net = copy.deepcopy(core_net(strategy.net)).cpu()
    loader_args = dict(strategy.train_args["loader_te_args"])
    with tempfile.TemporaryDirectory(prefix="al_shard_query_") as out_dir:
        mp.spawn(_shard_infer_worker,
                 args=(world, net, dataset, shards, loader_args, want_embedding,
                       out_dir),
                 nprocs=world, join=True)
        parts = [torch.load(os.path.join(out_dir, f"shard_{r}.pt"),
                            weights_only=False) for r in range(world)]
    logits = torch.cat([p["logits"] for p in parts], dim=0)
    emb = (torch.cat([p["emb"] for p in parts], dim=0)
           if parts[0]["emb"] is not None else None)
    labels = torch.cat([p["labels"] for p in parts], dim=0)
    if keep_device:
        logits = logits.to(strategy.device)
        if emb is not None:
            emb = emb.to(strategy.device)
    return logits, emb, labels
