"""Shared query-pass helpers: batched inference over pool subsets with
device-resident collection (the reference pages every batch back to CPU,
e.g. coreset_sampler.py:43-57; on MI355X the pool caches stay in HBM).

Multi-GPU query sharding: the reference's query pass is single-GPU by design
(every sampler forwards the whole pool on device 0 while the other GPUs sit
idle between training rounds). Here forward_pool shards the pool across all
visible GPUs whenever the strategy trains with world_size > 1 and the pool
is large enough (AL_SHARD_QUERY_MIN, default 4096). The workers are
PERSISTENT: spawned once per process lifetime, they cache the dataset and
net architecture, receive (weights, shard) tasks over torch mp queues and
return CPU tensors through shared memory — no per-query process spawn, no
temp-file gather. The parent computes shard 0 itself on its own device,
overlapping the workers. Kill switch: AL_SHARD_QUERY=0.
"""

import atexit
import copy
import os

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
    if want_embedding:
        fwd = lambda t: net(t, return_features="finalembed")  # noqa: E731
    else:
        fwd = net
    if strategy.device.type == "cuda" and os.environ.get("AL_EVAL_GRAPH", "1") == "1":
        from ..ops.graph import GraphedInference
        fwd = GraphedInference(fwd, strategy.device)
    logits_l, emb_l, y_l = [], [], []
    for x, y, _ in loader:
        if want_embedding:
            out, emb = fwd(x)
            emb_l.append(emb.float() if keep_device else emb.float().cpu())
        else:
            out = fwd(x)
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


def _infer_shard(net, device, dataset, shard, loader_args, want_embedding):
    loader = DataLoader(Subset(dataset, indices=list(shard)), shuffle=False,
                        drop_last=False, **loader_args)
    logits_l, emb_l, y_l = [], [], []
    with torch.no_grad():
        for x, y, _ in loader:
            x = x.to(device, non_blocking=True)
            if want_embedding:
                out, emb = net(x, return_features="finalembed")
                emb_l.append(emb.float())
            else:
                out = net(x)
            logits_l.append(out.float())
            y_l.append(y)
    return (torch.cat(logits_l, dim=0), torch.cat(emb_l, dim=0) if emb_l else None,
            torch.cat(y_l, dim=0))


def _pool_worker_loop(rank, task_q, res_q):
    """Persistent query worker: owns one GPU, caches dataset + net module
    across tasks; per task only the state_dict travels (shared-mem CPU
    tensors via the torch mp queue)."""
    if torch.cuda.is_available():
        device = torch.device("cuda", rank % torch.cuda.device_count())
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
    ds_cache = {}
    net_cache = {}
    while True:
        task = task_q.get()
        if task is None:
            return
        try:
            (ds_token, dataset, net_token, net, sd, shard, loader_args,
             want_embedding) = task
            if dataset is not None:
                ds_cache.clear()
                ds_cache[ds_token] = dataset
            if net is not None:
                net_cache.clear()
                net_cache[net_token] = net.to(device)
            model = net_cache[net_token]
            if sd is not None:
                model.load_state_dict(sd)
            model.to(device).eval()
            logits, emb, labels = _infer_shard(model, device, ds_cache[ds_token],
                                               shard, loader_args, want_embedding)
            res_q.put((rank, logits.cpu(), emb.cpu() if emb is not None else None,
                       labels, None))
        except Exception as e:  # surface to the parent, keep the worker alive
            import traceback
            res_q.put((rank, None, None, None, traceback.format_exc()))


class _QueryWorkerPool:
    """Lazy, process-lifetime pool of shard-inference workers (ranks 1..n-1;
    the parent is rank 0)."""

    def __init__(self):
        self.procs = []
        self.task_qs = []
        self.res_q = None
        self.n_workers = 0
        self._ds_token = 0
        self._ds_obj = None
        self._net_token = 0
        self._net_arch = None  # (class, repr of structure) proxy

    def ensure(self, n_workers):
        alive = self.procs and all(p.is_alive() for p in self.procs)
        if alive and self.n_workers == n_workers:
            return True
        self.shutdown()
        ctx = mp.get_context("spawn")
        self.res_q = ctx.Queue()
        self.task_qs = []
        self.procs = []
        for r in range(1, n_workers + 1):
            q = ctx.Queue()
            p = ctx.Process(target=_pool_worker_loop, args=(r, q, self.res_q),
                            daemon=True)
            p.start()
            self.task_qs.append(q)
            self.procs.append(p)
        self.n_workers = n_workers
        # new processes know nothing: force dataset+net resend
        self._ds_obj = None
        self._net_arch = None
        return True

    def shutdown(self):
        for q in self.task_qs:
            try:
                q.put(None)
            except Exception:
                pass
        for p in self.procs:
            p.join(timeout=5)
            if p.is_alive():
                p.terminate()
        self.procs, self.task_qs, self.res_q = [], [], None
        self.n_workers = 0

    def submit(self, dataset, net_cpu, shards, loader_args, want_embedding):
        """Send shards 1..n to the workers; returns the per-worker task count.
        Dataset and net module travel only when they changed."""
        if dataset is not self._ds_obj:
            self._ds_token += 1
            self._ds_obj = dataset
            send_ds = dataset
        else:
            send_ds = None
        arch = type(net_cpu).__name__ + str(sum(p.numel() for p in net_cpu.parameters()))
        if arch != self._net_arch:
            self._net_token += 1
            self._net_arch = arch
            send_net, send_sd = net_cpu, None
        else:
            send_net, send_sd = None, net_cpu.state_dict()
        for w, shard in enumerate(shards):
            self.task_qs[w].put((self._ds_token, send_ds, self._net_token,
                                 send_net, send_sd, shard, loader_args,
                                 want_embedding))
        return len(shards)

    def collect(self, n_tasks, timeout=900):
        out = {}
        for _ in range(n_tasks):
            rank, logits, emb, labels, err = self.res_q.get(timeout=timeout)
            if err is not None:
                raise RuntimeError(f"query worker {rank} failed:\n{err}")
            out[rank] = (logits, emb, labels)
        return out


_query_pool = _QueryWorkerPool()
atexit.register(_query_pool.shutdown)


def _sharded_forward_pool(strategy, dataset, idxs, want_embedding, keep_device):
    """Fan the pool forward pass out over every visible GPU via the
    persistent worker pool; the parent overlaps by computing shard 0 on its
    own device. Results concatenate in shard order (np.array_split preserves
    ordering), elementwise identical to the single-device path."""
    world = getattr(strategy, "world_size", 1) or 1
    if torch.cuda.is_available():
        world = min(world, torch.cuda.device_count())
    shards = [list(map(int, s)) for s in np.array_split(np.asarray(idxs), world)]
    loader_args = dict(strategy.train_args["loader_te_args"])

    net = core_net(strategy.net)
    net_cpu = copy.deepcopy(net).cpu()
    _query_pool.ensure(world - 1)
    n_tasks = _query_pool.submit(dataset, net_cpu, shards[1:], loader_args,
                                 want_embedding)

    net.eval()
    net.to(strategy.device)
    logits0, emb0, labels0 = _infer_shard(net, strategy.device, dataset,
                                          shards[0], loader_args, want_embedding)
    try:
        results = _query_pool.collect(n_tasks)
    except Exception:
        # a failed/timed-out worker may leave stale results in the queue;
        # tear the pool down so the next query starts clean
        _query_pool.shutdown()
        raise

    dev = strategy.device if keep_device else torch.device("cpu")
    logits_parts = [logits0.to(dev)]
    emb_parts = [emb0.to(dev)] if emb0 is not None else None
    label_parts = [labels0]
    for r in range(1, world):
        lg, em, lb = results[r]
        logits_parts.append(lg.to(dev))
        if emb_parts is not None:
            emb_parts.append(em.to(dev))
        label_parts.append(lb)
    logits = torch.cat(logits_parts, dim=0)
    emb = torch.cat(emb_parts, dim=0) if emb_parts is not None else None
    labels = torch.cat(label_parts, dim=0)
    if not keep_device:
        logits = logits.cpu()
        emb = emb.cpu() if emb is not None else None
    return logits, emb, labels
