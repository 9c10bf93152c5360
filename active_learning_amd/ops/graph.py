"""hipGraph capture of the training step for the REAL trainer.

The reference's metric is round wall-clock (main_al.py:160-178); the eager
epoch loop pays a launch gap per kernel (~200 launches/step on ResNet-50).
GraphedTrainStep captures forward + CE + backward + fused-SGD update into one
hipGraph after a few eager warmup steps and replays it per batch, with:

* static input/target buffers (H2D copy lands outside the graph);
* the LR schedule routed through a device hyper buffer
  (FusedSGD.enable_device_hyper) so scheduler.step() never forces re-capture;
* an eager fallback for odd-sized batches (the non-drop_last tail) and for
  anything that fails capture — capture failure can never lose training;
* post-replay cache upkeep: in-graph kernels update BN running stats and
  bf16 weight shadows in place without bumping the python-side version
  ticks, so replay bumps them (the folded-BN eval cache keys on _al_tick,
  ops/fused.py:26-43).

Single-process only: with BucketedDDP the RCCL all-reduce ordering is driven
by autograd hooks on the host, which a replay skips — world_size > 1 keeps
the eager loop (its collectives already overlap backward).
"""

import logging

import torch

from .functional import bump_tick

logger = logging.getLogger("ActiveLearning")


class GraphedInference:
    """hipGraph-captured no-grad forward for the query/eval loops: the pool
    pass replays one graph per batch instead of ~200 eager launches
    (reference: full-pool inference per query, e.g. confidence_sampler.py:
    27-36). Create one per pass — the folded-BN eval constants are baked at
    capture, so weights must not change across calls (true within a query
    or one validation sweep). Odd-sized tail batches fall back to eager."""

    def __init__(self, fn, device, warmup=1):
        self.fn = fn
        self.device = device
        self.warmup = warmup
        self._calls = 0
        self._graph = None
        self._failed = device.type != "cuda"
        self.x_static = None
        self.out_static = None

    @torch.no_grad()
    def __call__(self, x):
        if self._failed or (self.x_static is not None
                            and x.shape != self.x_static.shape):
            return self.fn(x.to(self.device, non_blocking=True))
        if self._graph is None:
            if self.x_static is None:
                self.x_static = torch.empty(x.shape, dtype=x.dtype,
                                            device=self.device)
            self.x_static.copy_(x)
            if self._calls < self.warmup:
                self._calls += 1
                return self.fn(self.x_static)
            try:
                torch.cuda.synchronize()
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    self.out_static = self.fn(self.x_static)
                self._graph = g
            except Exception as e:
                logger.warning(f"inference capture unavailable ({e!r}); eager")
                self._failed = True
                return self.fn(self.x_static)
        else:
            self.x_static.copy_(x, non_blocking=True)
        self._graph.replay()
        out = self.out_static
        if isinstance(out, (tuple, list)):
            return type(out)(o.clone() for o in out)
        return out.clone()


class GraphedTrainStep:
    """Wraps (net, optimizer, criterion) into a capturable step:
    loss = step(x, y). Capture happens lazily after `warmup` eager calls."""

    def __init__(self, net, optimizer, criterion, device, warmup=3):
        self.net = net
        self.opt = optimizer
        self.crit = criterion
        self.device = device
        self.warmup = warmup
        self._calls = 0
        self._graph = None
        self._failed = False
        self.x_static = None
        self.y_static = None
        self.loss_static = None
        self._bn_tensors = []
        for m in net.modules():
            for name in ("weight", "bias", "running_mean", "running_var"):
                t = getattr(m, name, None)
                if t is not None and hasattr(m, "running_mean"):
                    self._bn_tensors.append(t)
        can_hyper = hasattr(optimizer, "enable_device_hyper") and \
            len(optimizer.param_groups) == 1
        self._persistent = False
        if can_hyper and device.type == "cuda":
            optimizer.enable_device_hyper(device)
            # persistent grads: storage stays put across steps (backward
            # accumulates in place, the fused SGD zeroes after use) so the
            # captured step is allocation-free and the chunk table is stable
            optimizer.enable_persistent_grads()
            self._persistent = True

    # ------------------------------------------------------------------ #
    def _dealias_grads(self):
        """Persistent grads must not ALIAS the conv-wgrad arena: backward
        writes dw into the arena view and then accumulates p.grad += dw —
        if p.grad IS the view, the grad doubles. Clone any aliased grad into
        its own stable storage (runs between eager steps, outside capture)."""
        from .functional import _grad_arena
        flat = _grad_arena.flat
        if flat is None:
            return
        lo = flat.data_ptr()
        hi = lo + flat.numel() * flat.element_size()
        for group in self.opt.param_groups:
            for p in group["params"]:
                g = p.grad
                if g is not None and g.is_cuda and lo <= g.data_ptr() < hi:
                    p.grad = g.detach().clone()

    def _eager(self, x, y):
        if getattr(self.opt, "_hyper_dev", None) is not None:
            self.opt.sync_hyper()
        x = x.to(self.device, non_blocking=True)
        y = y.to(self.device, non_blocking=True)
        if not self._persistent:
            self.opt.zero_grad(set_to_none=True)
        out = self.net(x)
        loss = self.crit(out, y)
        loss.backward()
        if self._persistent:
            self._dealias_grads()
        self.opt.step()
        # return DETACHED: a caller-held loss would keep this step's autograd
        # graph (incl. default-stream AccumulateGrad nodes) alive across the
        # upcoming capture, which aborts hipGraph capture with a stream
        # mismatch (observed: core dump in torch 2.10's input_buffer.cpp)
        return loss.detach()

    def _capture(self):
        import gc
        gc.collect()  # drop any lingering warmup autograd graphs for certain
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            out = self.net(self.x_static)
            loss = self.crit(out, self.y_static)
            loss.backward()
            self.opt.step()
            # detached view shares storage: each replay refreshes the value
            self.loss_static = loss.detach()
        self._graph = g

    def _post_replay(self):
        # in-graph kernels updated BN stats + weight shadows in place; bump
        # the python-side ticks so folded-eval caches refold (fused.py)
        for t in self._bn_tensors:
            bump_tick(t)

    # ------------------------------------------------------------------ #
    def step(self, x, y):
        """x, y on CPU (pinned loader) or device. Returns the loss tensor."""
        if self._failed or self.device.type != "cuda":
            return self._eager(x, y)
        if self.x_static is not None and x.shape != self.x_static.shape:
            return self._eager(x, y)  # tail batch
        if self._graph is None:
            if self.x_static is None:
                self.x_static = torch.empty(x.shape, dtype=x.dtype,
                                            device=self.device)
                self.y_static = torch.empty(y.shape, dtype=y.dtype,
                                            device=self.device)
            self.x_static.copy_(x)
            self.y_static.copy_(y)
            if self._calls < self.warmup:
                self._calls += 1
                return self._eager(self.x_static, self.y_static)
            try:
                torch.cuda.synchronize()
                self._capture()
                logger.info("hipGraph-captured the training step "
                            f"(batch {tuple(x.shape)})")
            except Exception as e:
                logger.warning(f"hipGraph capture unavailable ({e!r}); "
                               "training eagerly")
                self._failed = True
                self._graph = None
                return self._eager(self.x_static, self.y_static)
            self._graph.replay()
            self._post_replay()
            return self.loss_static
        if getattr(self.opt, "_hyper_dev", None) is not None:
            self.opt.sync_hyper()  # track the LR schedule between replays
        self.x_static.copy_(x, non_blocking=True)
        self.y_static.copy_(y, non_blocking=True)
        self._graph.replay()
        self._post_replay()
        return self.loss_static
