"""Strategy base class: pool state + distributed trainer.

Structural parity with src/query_strategies/strategy.py (the reference's base
class IS the trainer: pool masks, DDP launch, epoch loop, early stopping,
best-checkpoint bookkeeping). MI355X-native differences:

* collectives via RCCL over xGMI through torch.distributed ("nccl" backend on
  ROCm; "gloo" for CPU tests) with a first-party BucketedDDP whose gradient
  all-reduce overlaps backward (parallel/ddp.py);
* optimizers/schedulers/criterion come from explicit registries
  (ops/optim.py, ops/loss.py) instead of eval() of config strings
  (strategy.py:345-350);
* the model runs NHWC/bf16 on hand-written HIP kernels (ops/).

Pool-state and checkpoint semantics preserved (file:line cites per method).
"""

import logging
import os

import numpy as np
import torch
import torch.multiprocessing as mp
import torch.distributed as dist
from torch.utils.data import DataLoader, Subset

from ..models.init import init_params
from ..ops.loss import CrossEntropyLoss
from ..ops.optim import build_optimizer, build_scheduler
from ..parallel import BucketedDDP, convert_sync_batchnorm, get_free_tcp_port
from ..utils.checkpoint import load_pretrained_weights, state_dict_with_marker
from ..utils.evaluation import evaluate, gather_parallel_eval


class Strategy:
    """Base class for an active-learning query strategy (and the trainer).

    Attributes mirror the reference (strategy.py:21-72): train_set / al_set /
    test_set, idxs_lb / idxs_lb_recent boolean masks over n_pool, eval_idxs,
    cumulative_cost, round, es_params, world_size, net, freeze_feature.
    """

    logger = logging.getLogger("ActiveLearning")

    def __init__(self, train_set, al_set, net, train_args, eval_idxs, comet_experiment,
                 test_set=None, **kwargs):
        self.train_args = train_args
        self.comet_experiment = comet_experiment
        self.train_set = train_set
        self.al_set = al_set
        self.test_set = test_set

        self.num_classes = self.al_set.num_classes
        self.logger.info(f"Number of classes: {self.num_classes}")

        self.round = 0
        self.cumulative_cost = 0

        self.n_pool = len(self.al_set)
        self.eval_idxs = eval_idxs
        self.idxs_lb = np.zeros(self.n_pool, dtype=bool)
        self.idxs_lb_recent = np.zeros(self.n_pool, dtype=bool)

        self.es_params = {"use_es": kwargs["early_stop_patience"] != 0,
                          "patience": kwargs["early_stop_patience"], "count": 0,
                          "success": False, "best_perf": 0}
        self.n_epoch = kwargs["n_epoch"]
        self.imbalanced_training = train_args.get("imbalanced_training", False)

        self.world_size = kwargs.get("world_size") or 1
        self.backend = kwargs.get("backend")  # None -> auto (nccl on GPU)

        use_cuda = torch.cuda.is_available()
        self.device = torch.device("cuda" if use_cuda else "cpu")
        self.logger.info(f"Using device: {self.device}")
        self.net = net
        self.net_name = kwargs["model"]
        self.query_net = None
        self.feature_net = net
        self.freeze_feature = kwargs["freeze_feature"]
        self.best_epoch = 0

        self.base_ckpt_path = kwargs["ckpt_path"]
        self.exp_name = kwargs["exp_name"]
        self.exp_hash = kwargs.get("exp_hash") or "no_comet"

        # strategy-specific kwargs used by subclasses
        self._kwargs = {k: v for k, v in kwargs.items()
                        if k in ("subset_labeled", "subset_unlabeled", "partitions",
                                 "vae_latent_dim", "vaal_adversary_param", "lr_vae",
                                 "lr_discriminator")}

    # ------------------------------------------------------------------ #
    # pool bookkeeping
    # ------------------------------------------------------------------ #

    def available_query_idxs(self, boolean=False, shuffle=True):
        """Idxs not yet labeled and not in the eval split (strategy.py:126-145)."""
        if boolean:
            mask = ~self.idxs_lb
            mask[self.eval_idxs] = False
            return mask
        mask = ~self.idxs_lb
        mask[self.eval_idxs] = False
        idxs = np.where(mask)[0]
        if shuffle:
            idxs = np.random.permutation(idxs)
        return idxs

    def already_labeled_idxs(self, boolean=False, shuffle=False):
        if boolean:
            return np.copy(self.idxs_lb)
        idxs = np.where(self.idxs_lb)[0]
        if shuffle:
            idxs = np.random.permutation(idxs)
        return idxs

    def update(self, labeled_idxs, cur_cost):
        """Mark newly queried samples; assert no double labeling
        (strategy.py:459-484)."""
        if isinstance(labeled_idxs, list):
            labeled_idxs = np.array(labeled_idxs)
        self.idxs_lb_recent = labeled_idxs
        for idx in np.asarray(labeled_idxs, dtype=np.int64).ravel():
            assert not self.idxs_lb[idx], f"sample {idx} already labeled"
            self.idxs_lb[idx] = True
        self.cumulative_cost += cur_cost
        self.comet_experiment.log_metric("cumulative_budget", self.cumulative_cost,
                                         include_context=False, step=self.round)
        self.logger.info(f"Cumulative budget used on round {self.round} = "
                         f"{self.cumulative_cost}")
        self.comet_experiment.log_asset_data(
            ",".join(str(e) for e in np.asarray(labeled_idxs).ravel()),
            name=f"labeled_idxs_on_rd_{self.round}")
        out_dir = os.path.join(self.base_ckpt_path, self.exp_name)
        os.makedirs(out_dir, exist_ok=True)
        with open(os.path.join(out_dir, "labeled_idxs_per_round.txt"), "a") as fh:
            fh.writelines(f"Round {self.round}: {labeled_idxs}\n")

    # ------------------------------------------------------------------ #
    # weights
    # ------------------------------------------------------------------ #

    def generate_weight_paths(self):
        ckpt_dir = os.path.join(self.base_ckpt_path, f"{self.exp_name}_{self.exp_hash}")
        os.makedirs(ckpt_dir, exist_ok=True)
        return {"best_ckpt": os.path.join(ckpt_dir, f"best_rd_{self.round}.pth"),
                "previous_ckpt": os.path.join(ckpt_dir, f"rd_{self.round - 1}.pth"),
                "current_ckpt": os.path.join(ckpt_dir, f"rd_{self.round}.pth")}

    def init_network_weights(self):
        """Random re-init each round; then optionally load SSL/transfer ckpt
        with key surgery (strategy.py:175-199)."""
        init_ckpt_path = self.train_args.get("init_pretrained_ckpt_path")
        init_params(self.net)  # reference: net.apply(init_params), strategy.py:184
        if init_ckpt_path is None:
            self.logger.info("Initialized Network Weights Randomly.")
        else:
            self.logger.info(f"Initializing Network Weights from {init_ckpt_path}")
            self.net = load_pretrained_weights(
                self.net, init_ckpt_path,
                replace_key=self.train_args.get("replace_key"),
                skip_key=self.train_args.get("skip_key"),
                required_key=self.train_args.get("required_key"))
        self.feature_net = self.net

    def load_best_ckpt(self):
        best = self.generate_weight_paths()["best_ckpt"]
        self.logger.info(f"Loading best ckpt so far from: {best}")
        self.net = load_pretrained_weights(self.net, best)

    # ------------------------------------------------------------------ #
    # query / test
    # ------------------------------------------------------------------ #

    def query(self, budget):
        raise NotImplementedError

    def test(self):
        """Single-process full test-set evaluation (strategy.py:211-247)."""
        if not self.test_set:
            self.logger.info("Skipped testing loop, no testing dataset found.")
            return None
        self.net.to(self.device)
        loader_te_args = dict(self.train_args["loader_te_args"])
        loader_te_args["batch_size"] = max(1, int(loader_te_args["batch_size"]
                                                  / self.world_size))
        test_loader = DataLoader(self.test_set, shuffle=False, **loader_te_args,
                                 drop_last=False, pin_memory=self.device.type == "cuda")
        perf = evaluate(test_loader, net=self.net, metric="accuracy",
                        num_classes=self.num_classes, net_name=self.net_name)
        test_perf = perf["accuracy"].cpu()
        test_top5 = perf["top_5_accuracy"].cpu()
        byclass = perf["accuracy_byclass"].cpu()
        order = sorted(range(len(byclass)), key=lambda k: byclass[k])
        tmp = int(min(5, len(byclass)))
        best_c = {i: f"{byclass[i].item() * 100:.2f}" for i in order[-tmp:]}
        worst_c = {i: f"{byclass[i].item() * 100:.2f}" for i in order[:tmp]}
        self.logger.info(f"Test performance at round {self.round} is "
                         f"{test_perf * 100:.2f}%")
        self.logger.info(f"Best {tmp} classes: {best_c}; worst {tmp}: {worst_c}")
        self.logger.info(f"Test top 5 acc at round {self.round} is {test_top5 * 100:.2f}%")
        self.comet_experiment.log_metrics(
            {"rd_test_accuracy": test_perf, "rd_test_top5_accuracy": test_top5},
            step=self.round)
        self.comet_experiment.log_metrics(
            {"budget_test_accuracy": test_perf, "budget_test_top5_accuracy": test_top5},
            step=self.cumulative_cost)
        self.comet_experiment.log_asset_data(
            ",".join(f"{e.item():.2f}" for e in byclass),
            name=f"test_acc_byclass_rd_{self.round}")
        return test_perf

    # ------------------------------------------------------------------ #
    # training
    # ------------------------------------------------------------------ #

    def generate_imbalanced_training_weights(self):
        """Inverse-frequency class weights (strategy.py:444-457)."""
        idxs = self.already_labeled_idxs(boolean=False, shuffle=False)
        labels, counts = np.unique(np.asarray(self.train_set.targets)[idxs],
                                   return_counts=True)
        weights = np.ones(self.num_classes)
        total = counts.sum()
        for i, lbl in enumerate(labels):
            weights[lbl] = total / counts[i]
        weights /= weights.sum()
        return torch.tensor(weights, dtype=torch.float32)

    def train(self):
        """Per-round (re)spawn of one process per GPU (strategy.py:286-302)."""
        self.imb_weights = self.generate_imbalanced_training_weights()
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(get_free_tcp_port())
        if self.world_size > 1:
            convert_sync_batchnorm(self.net)
            tmp = self.comet_experiment
            self.comet_exp_key = self.comet_experiment.get_key()
            self.comet_experiment = None
            mp.spawn(self._spawn_entry, args=(), nprocs=self.world_size, join=True)
            self.comet_experiment = tmp
            convert_sync_batchnorm(self.net, process_group=False)
            # results come back via checkpoint files (rank 0 writes; parent
            # reloads through load_best_ckpt, strategy.py:214-217 semantics)
        else:
            self.parallel_train_fn(0)

    def _spawn_entry(self, rank):
        try:
            self.parallel_train_fn(rank)
        finally:
            if dist.is_initialized():
                dist.destroy_process_group()

    # -- per-rank training fn (strategy.py:304-381) ----------------------- #

    def _init_distributed(self, rank):
        if self.world_size > 1:
            backend = self.backend or ("nccl" if torch.cuda.is_available() else "gloo")
            dist.init_process_group(backend, rank=rank, world_size=self.world_size)
            if torch.cuda.is_available():
                torch.cuda.set_device(rank)

    def _rank_device(self, rank):
        if torch.cuda.is_available():
            return torch.device("cuda", rank if self.world_size > 1 else
                                torch.cuda.current_device())
        return torch.device("cpu")

    def build_train_objects(self, net):
        optimizer = build_optimizer(self.train_args["optimizer"], net.parameters(),
                                    **self.train_args["optimizer_args"])
        scheduler = build_scheduler(self.train_args["lr_scheduler"], optimizer,
                                    **self.train_args["lr_scheduler_args"])
        weight = self.imb_weights if self.imbalanced_training else None
        criterion = CrossEntropyLoss(weight=weight)
        return optimizer, scheduler, criterion

    def parallel_train_fn(self, rank):
        weight_paths = self.generate_weight_paths()
        train_subset = Subset(self.train_set,
                              self.already_labeled_idxs(boolean=False, shuffle=False))
        self._init_distributed(rank)
        device = self._rank_device(rank)
        self.device = device

        if self.world_size > 1:
            train_sampler = torch.utils.data.distributed.DistributedSampler(
                train_subset, num_replicas=self.world_size, rank=rank, shuffle=True)
            if rank == 0 and getattr(self, "comet_exp_key", None):
                from ..utils.tracking import ExistingExperiment
                self.comet_experiment = ExistingExperiment(
                    previous_experiment=self.comet_exp_key)
        else:
            train_sampler = None

        loader_tr_args = dict(self.train_args["loader_tr_args"])
        loader_tr_args["batch_size"] = max(1, int(loader_tr_args["batch_size"]
                                                  / self.world_size))
        loader_tr = DataLoader(train_subset, shuffle=(train_sampler is None),
                               **loader_tr_args, drop_last=False, sampler=train_sampler,
                               pin_memory=device.type == "cuda")

        self.net = self.net.to(device)
        self.net.train()
        if self.world_size > 1:
            self.net = BucketedDDP(self.net)

        self.es_params.update(count=0, success=False, best_perf=0)
        step = 0

        optimizer, scheduler, criterion = self.build_train_objects(self.net)
        criterion = criterion.to(device)
        print(f"Rank {rank} training starts.")
        self.logger.info(f"Starting training on round {self.round}")

        # hipGraph-capture the train step (fwd+CE+bwd+fused SGD in one
        # replay, zero launch gaps). Single-process only: the DDP all-reduce
        # schedule is host-hook-driven. AL_TRAIN_GRAPH=0 disables.
        graphed = None
        if (self.world_size == 1 and device.type == "cuda"
                and os.environ.get("AL_TRAIN_GRAPH", "1") == "1"):
            from ..ops.graph import GraphedTrainStep
            graphed = GraphedTrainStep(self.net, optimizer, criterion, device)

        # subclasses that co-train auxiliary models (VAAL) hook in here
        self._setup_aux(device, rank)

        for epoch in range(1, self.n_epoch + 1):
            if train_sampler is not None:
                train_sampler.set_epoch(epoch)
            self.net.train()
            # BN-freeze semantics when linear-probing / finetuning a
            # pretrained backbone (strategy.py:363-367)
            if self.freeze_feature or ("init_pretrained_ckpt_path" in self.train_args):
                self.net.eval()
            self._epoch_start(epoch)
            step = self._train(rank=rank, epoch=epoch, loader_tr=loader_tr,
                               optimizer=optimizer, criterion=criterion, step=step,
                               graphed=graphed)
            scheduler.step()
            self._epoch_end()
            if self.validation_and_early_stopping(rank, epoch, weight_paths):
                break

        msg = f"Sanity Check: Best ckpt of worker rank {rank} occurs on epoch " \
              f"{self.best_epoch}"
        print(msg)
        self.logger.info(msg)
        self.logger.info(f"Finished training on round {self.round}")
        self._teardown_aux()
        if self.world_size > 1 and isinstance(self.net, BucketedDDP):
            self.net = self.net.module

    # -- trainer hooks for strategies that co-train auxiliary models -------- #

    def _setup_aux(self, device, rank):
        """Called once per round after the classifier is on-device/DDP-wrapped
        and its optimizer exists; VAAL builds its VAE/discriminator stack
        here."""

    def _epoch_start(self, epoch):
        """Called at each epoch start after the classifier's train/eval-mode
        bookkeeping."""

    def _epoch_end(self):
        """Called after the classifier's scheduler steps."""

    def _teardown_aux(self):
        """Called when the round's training finishes (before DDP unwrap)."""

    def _train(self, rank, epoch, loader_tr, optimizer, criterion, step,
               graphed=None):
        """One training epoch — the hot loop (strategy.py:249-284). The loss
        stays on-device between log points: a per-batch .cpu() would force a
        full pipeline sync every iteration."""
        for batch_idx, (x, y, _idxs) in enumerate(loader_tr):
            if graphed is not None:
                loss = graphed.step(x, y)
                self._log_train_batch(rank, epoch, batch_idx, loader_tr, loss)
                step += 1
                continue
            x = x.to(self.device, non_blocking=True)
            y = y.to(self.device, non_blocking=True)
            optimizer.zero_grad(set_to_none=True)
            out = self.net(x)
            loss = criterion(out, y)
            loss.backward()
            if isinstance(self.net, BucketedDDP):
                self.net.finalize_grads()
            optimizer.step()
            self._log_train_batch(rank, epoch, batch_idx, loader_tr, loss)
            step += 1
        return step

    def _log_train_batch(self, rank, epoch, batch_idx, loader_tr, loss):
        if batch_idx % 25 == 0:  # reference cadence (strategy.py:276-279)
            cur_loss = loss.detach().float().cpu()
            msg = (f"\tRound {self.round}, Epoch {epoch}, batch "
                   f"{batch_idx}/{len(loader_tr)}, loss is {cur_loss} on worker "
                   f"rank {rank}")
            self.logger.info(msg)
            if self.world_size == 1 or rank == 1:
                print(msg)

    # -- validation / early stop (strategy.py:383-442) --------------------- #

    def validation_and_early_stopping(self, rank, epoch, weight_paths):
        if not self.es_params["use_es"] or len(self.eval_idxs) == 0:
            # no validation data (early stop off, or the balanced eval split
            # collapsed to zero on a tiny pool): keep checkpoints current
            if rank == 0:
                torch.save(state_dict_with_marker(self.net), weight_paths["best_ckpt"])
                torch.save(state_dict_with_marker(self.net),
                           weight_paths["current_ckpt"])
            return False
        validation_data = Subset(self.al_set, indices=self.eval_idxs)
        if self.world_size > 1:
            val_sampler = torch.utils.data.distributed.DistributedSampler(
                validation_data, num_replicas=self.world_size, rank=rank, shuffle=False)
        else:
            val_sampler = None
        loader_te_args = dict(self.train_args["loader_te_args"])
        loader_te_args["batch_size"] = max(1, int(loader_te_args["batch_size"]
                                                  / self.world_size))
        loader = DataLoader(validation_data, shuffle=False, **loader_te_args,
                            drop_last=False, sampler=val_sampler,
                            pin_memory=self.device.type == "cuda")
        perf = evaluate(loader, net=self.net, metric="accuracy",
                        num_classes=self.num_classes, net_name=self.net_name)
        if self.world_size > 1:
            # NOTE: with a DistributedSampler each rank sees a shard but
            # `count`/accuracy denominators use len(dataset); the all-reduced
            # counts are correct because shards partition the dataset (padding
            # duplicates at most world_size-1 samples, as in the reference).
            perf["count"] = len(val_sampler)
            eval_perf, eval_top5, _byclass = gather_parallel_eval(
                perf, self.world_size, self.device)
        else:
            eval_perf = perf["accuracy"].cpu()
            eval_top5 = perf["top_5_accuracy"].cpu()

        if self.world_size == 1 or rank == 1:
            msg = (f"\tValidation performance on round {self.round} at epoch {epoch} "
                   f"is {eval_perf * 100:.2f}%")
            self.logger.info(msg)
            print(msg)
            self.logger.info(f"\tValidation top5 acc on round {self.round} at epoch "
                             f"{epoch} is {eval_top5 * 100:.2f}%")
        if epoch % 25 == 0 and rank == 0 and self.comet_experiment is not None:
            self.comet_experiment.log_metrics(
                {f"rd_{self.round}_validation_accuracy": eval_perf,
                 f"rd_{self.round}_validation_top5_accuracy": eval_top5}, step=epoch)

        if eval_perf >= self.es_params["best_perf"]:
            self.best_epoch = epoch
            self.es_params["count"] = 0
            self.es_params["best_perf"] = eval_perf
            if rank == 0:
                torch.save(state_dict_with_marker(self.net), weight_paths["best_ckpt"])
        else:
            self.es_params["count"] += 1

        if self.es_params["count"] > self.es_params["patience"]:
            if rank == 0:
                self.logger.info("Early stopping criterion reached.")
            return True
        if rank == 0:
            torch.save(state_dict_with_marker(self.net), weight_paths["current_ckpt"])
        return False

    # ------------------------------------------------------------------ #
    # helpers shared by samplers
    # ------------------------------------------------------------------ #

    def _eval_loader(self, dataset, idxs=None):
        ds = dataset if idxs is None else Subset(dataset, indices=idxs)
        return DataLoader(ds, shuffle=False, **self.train_args["loader_te_args"],
                          drop_last=False)

    def __getstate__(self):
        state = dict(self.__dict__)
        # never pickle a live DDP wrapper
        net = state.get("net")
        if isinstance(net, BucketedDDP):
            state["net"] = net.module
        return state
