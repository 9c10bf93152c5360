"""VAAL: task-agnostic AL with a co-trained VAE + discriminator.

Reference: src/query_strategies/vaal_sampler.py — latent_scale 1 for 10
classes / 2 for 1000 (:24-29); query scores the pool with the discriminator
on the VAE mu and takes the most-unlabeled-looking (:39-70); overridden
parallel_train_fn co-trains classifier + VAE + discriminator with three
optimizers (SGD + 2x Adam, :134-140); per batch: classifier step (:216-220),
VAE step = labeled recon + transductive recon + adversarial BCE with
--vaal_adversary_param (:230-248), discriminator step labeled=1/unlabeled=0
(:250-268); vae_loss = MSE + beta*KLD (:276-280).
"""

import numpy as np
import torch
import torch.nn as nn
from torch.utils.data import DataLoader, Subset

from ..models.vae import VAE, Discriminator
from ..ops.optim import FusedAdam
from ..parallel import BucketedDDP
from ..utils.checkpoint import state_dict_with_marker
from .strategy import Strategy


class VAALSampler(Strategy):
    def __init__(self, train_set, al_set, net, train_args, eval_idxs, comet_experiment,
                 test_set=None, **kwargs):
        super().__init__(train_set, al_set, net, train_args, eval_idxs,
                         comet_experiment, test_set, **kwargs)
        if self.num_classes == 10:
            self.latent_scale = 1.0
        elif self.num_classes == 1000:
            self.latent_scale = 2.0
        else:
            raise ValueError("Unsupported dataset")
        self.vae = VAE(z_dim=kwargs["vae_latent_dim"], nc=3,
                       latent_scale=self.latent_scale)
        self.discriminator = Discriminator(z_dim=kwargs["vae_latent_dim"])
        self.bce_loss = nn.BCELoss()
        self.mse_loss = nn.MSELoss()
        self.adversary_param = kwargs["vaal_adversary_param"]
        self.lr_discriminator = kwargs["lr_discriminator"]
        self.lr_vae = kwargs["lr_vae"]

    # -- query (vaal_sampler.py:39-70) ------------------------------------- #
    @torch.no_grad()
    def query(self, budget):
        idxs_for_query = self.available_query_idxs()
        loader = DataLoader(Subset(self.al_set, idxs_for_query), shuffle=False,
                            **self.train_args["loader_te_args"], drop_last=False)
        self.net.eval()
        self.vae.eval()
        self.discriminator.eval()
        self.vae.to(self.device)
        self.discriminator.to(self.device)

        all_preds, all_indices = [], []
        for x, _y, idxs in loader:
            x = x.to(self.device)
            _, _, _, mu, _ = self.vae(x)
            preds = self.discriminator(mu)
            all_preds.append(preds.view(-1).cpu())
            all_indices.extend(idxs.tolist())
        all_preds = torch.cat(all_preds) * -1  # most-unlabeled-looking first
        query_count = int(min(len(idxs_for_query), budget))
        _, top = torch.topk(all_preds, query_count)
        labeled_idxs = np.asarray(all_indices)[top.numpy()]
        return labeled_idxs, query_count

    def init_network_weights(self):
        super().init_network_weights()
        self.vae.weight_init()
        self.discriminator.weight_init()

    # -- co-training (vaal_sampler.py:77-183) ------------------------------ #
    def parallel_train_fn(self, rank):
        weight_paths = self.generate_weight_paths()
        train_subset = Subset(self.train_set,
                              self.already_labeled_idxs(boolean=False, shuffle=False))
        unlabeled_subset = Subset(self.train_set,
                                  self.available_query_idxs(boolean=False, shuffle=False))
        self._init_distributed(rank)
        device = self._rank_device(rank)
        self.device = device

        if self.world_size > 1:
            train_sampler = torch.utils.data.distributed.DistributedSampler(
                train_subset, num_replicas=self.world_size, rank=rank, shuffle=True)
            unlabeled_sampler = torch.utils.data.distributed.DistributedSampler(
                unlabeled_subset, num_replicas=self.world_size, rank=rank, shuffle=True)
            if rank == 0 and getattr(self, "comet_exp_key", None):
                from ..utils.tracking import ExistingExperiment
                self.comet_experiment = ExistingExperiment(
                    previous_experiment=self.comet_exp_key)
        else:
            train_sampler = unlabeled_sampler = None

        loader_tr_args = dict(self.train_args["loader_tr_args"])
        loader_tr_args["batch_size"] = max(1, int(loader_tr_args["batch_size"]
                                                  / self.world_size))
        loader_tr = DataLoader(train_subset, shuffle=(train_sampler is None),
                               **loader_tr_args, drop_last=False, sampler=train_sampler,
                               pin_memory=device.type == "cuda")
        loader_unlabeled = DataLoader(unlabeled_subset,
                                      shuffle=(unlabeled_sampler is None),
                                      **loader_tr_args, drop_last=False,
                                      sampler=unlabeled_sampler,
                                      pin_memory=device.type == "cuda")

        self.net = self.net.to(device)
        self.vae = self.vae.to(device)
        self.discriminator = self.discriminator.to(device)
        self.net.train()
        self.vae.train()
        self.discriminator.train()
        if self.world_size > 1:
            self.net = BucketedDDP(self.net)
            self.vae = BucketedDDP(self.vae)
            self.discriminator = BucketedDDP(self.discriminator)

        self.es_params.update(count=0, success=False, best_perf=0)
        step = 0
        optimizer, scheduler, criterion = self.build_train_objects(self.net)
        criterion = criterion.to(device)
        optim_vae = FusedAdam(self.vae.parameters(), lr=self.lr_vae)
        optim_disc = FusedAdam(self.discriminator.parameters(), lr=self.lr_discriminator)
        from ..ops.optim import build_scheduler
        sched_vae = build_scheduler(self.train_args["lr_scheduler"], optim_vae,
                                    **self.train_args["lr_scheduler_args"])
        sched_disc = build_scheduler(self.train_args["lr_scheduler"], optim_disc,
                                     **self.train_args["lr_scheduler_args"])

        print(f"Rank {rank} training starts.")
        self.logger.info(f"Starting training on round {self.round}")
        for epoch in range(1, self.n_epoch + 1):
            if train_sampler is not None:
                train_sampler.set_epoch(epoch)
                unlabeled_sampler.set_epoch(epoch)
            self.net.train()
            self.vae.train()
            self.discriminator.train()
            if self.freeze_feature or ("init_pretrained_ckpt_path" in self.train_args):
                self.net.eval()
            step = self.vaal_train(rank=rank, epoch=epoch, loader_tr=loader_tr,
                                   optimizer=optimizer, criterion=criterion, step=step,
                                   loader_unlabeled_data=loader_unlabeled,
                                   optim_vae=optim_vae, optim_discriminator=optim_disc)
            scheduler.step()
            sched_vae.step()
            sched_disc.step()
            if self.validation_and_early_stopping(rank, epoch, weight_paths):
                break
        self.logger.info(f"Finished training on round {self.round}")
        for attr in ("net", "vae", "discriminator"):
            m = getattr(self, attr)
            if isinstance(m, BucketedDDP):
                setattr(self, attr, m.module)

    def vaal_train(self, rank, epoch, loader_tr, optimizer, criterion, step,
                   loader_unlabeled_data, optim_vae, optim_discriminator):
        unlabeled_iter = iter(loader_unlabeled_data)
        vae_core = self.vae.module if isinstance(self.vae, BucketedDDP) else self.vae
        for batch_idx, (x, y, _idxs) in enumerate(loader_tr):
            x = x.to(self.device, non_blocking=True)
            y = y.to(self.device, non_blocking=True)
            try:
                x_u, _, _ = next(unlabeled_iter)
            except StopIteration:
                unlabeled_iter = iter(loader_unlabeled_data)
                x_u, _, _ = next(unlabeled_iter)
            x_u = x_u.to(self.device, non_blocking=True)

            vae_core.set_crop_seed(int(np.random.randint(0, 10000)))

            # classifier step
            optimizer.zero_grad(set_to_none=True)
            out = self.net(x)
            loss = criterion(out, y)
            loss.backward()
            if isinstance(self.net, BucketedDDP):
                self.net.finalize_grads()
            optimizer.step()
            if batch_idx % 25 == 0:
                msg = (f"\tRound {self.round}, Epoch {epoch}, batch "
                       f"{batch_idx}/{len(loader_tr)}, loss is "
                       f"{loss.detach().float().cpu()} on worker rank {rank}")
                self.logger.info(msg)
                if self.world_size == 1 or rank == 1:
                    print(msg)

            # VAE step
            x_crop, recon, z, mu, logvar = self.vae(x)
            unsup_loss = self.vae_loss(x_crop, recon, mu, logvar, 1)
            xu_crop, u_recon, _uz, u_mu, u_logvar = self.vae(x_u)
            transductive_loss = self.vae_loss(xu_crop, u_recon, u_mu, u_logvar, 1)
            labeled_preds = self.discriminator(mu)
            unlabeled_preds = self.discriminator(u_mu)
            ones_l = torch.ones(x.size(0), device=self.device)
            ones_u = torch.ones(x_u.size(0), device=self.device)
            dsc_loss = (self.bce_loss(labeled_preds.squeeze(1), ones_l)
                        + self.bce_loss(unlabeled_preds.squeeze(1), ones_u))
            total_vae_loss = unsup_loss + transductive_loss \
                + self.adversary_param * dsc_loss
            optim_vae.zero_grad(set_to_none=True)
            total_vae_loss.backward()
            if isinstance(self.vae, BucketedDDP):
                self.vae.finalize_grads()
            optim_vae.step()

            # discriminator step
            with torch.no_grad():
                _, _, _, mu, _ = self.vae(x)
                _, _, _, u_mu, _ = self.vae(x_u)
            labeled_preds = self.discriminator(mu)
            unlabeled_preds = self.discriminator(u_mu)
            zeros_u = torch.zeros(x_u.size(0), device=self.device)
            dsc_loss = (self.bce_loss(labeled_preds.squeeze(1), ones_l)
                        + self.bce_loss(unlabeled_preds.squeeze(1), zeros_u))
            optim_discriminator.zero_grad(set_to_none=True)
            dsc_loss.backward()
            if isinstance(self.discriminator, BucketedDDP):
                self.discriminator.finalize_grads()
            optim_discriminator.step()
            step += 1
        return step

    def vae_loss(self, x, recon, mu, logvar, beta):
        mse = self.mse_loss(recon, x)
        kld = -0.5 * torch.sum(1 + logvar - mu.pow(2) - logvar.exp()) * beta
        return mse + kld
