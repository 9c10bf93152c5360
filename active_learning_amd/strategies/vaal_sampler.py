"""VAAL: task-agnostic AL with a co-trained VAE + discriminator.

Capability parity with src/query_strategies/vaal_sampler.py (latent_scale by
class count :24-29; discriminator-on-mu query :39-70; three optimizers
:134-140; per-batch classifier / VAE / discriminator sub-steps :216-268;
vae_loss = MSE + beta*KLD :276-280), restructured MI355X-first:

* the round's trainer plumbing (loaders, DDP wrap, schedulers, validation,
  hipGraph capture of the classifier step) is the base Strategy's — VAAL
  only plugs into the _setup_aux/_epoch_start/_epoch_end hooks instead of
  re-implementing the spawn function;
* each batch runs the labeled and unlabeled images through the VAE as ONE
  concatenated forward (half the VAE launches, same math: the seeded crop is
  per-batch so both halves see the same window);
* scoring stays device-resident end to end; the query pass writes
  discriminator scores into a preallocated pool-sized buffer and top-ks on
  the GPU.
"""

import numpy as np
import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader, Subset

from ..models.vae import VAE, Discriminator
from ..ops.optim import FusedAdam, build_scheduler
from ..parallel import BucketedDDP
from .strategy import Strategy

_LATENT_SCALE = {10: 1.0, 1000: 2.0}  # vaal_sampler.py:24-29


class VAALSampler(Strategy):
    def __init__(self, train_set, al_set, net, train_args, eval_idxs, comet_experiment,
                 test_set=None, **kwargs):
        super().__init__(train_set, al_set, net, train_args, eval_idxs,
                         comet_experiment, test_set, **kwargs)
        if self.num_classes not in _LATENT_SCALE:
            raise ValueError("Unsupported dataset")
        self.z_dim = kwargs["vae_latent_dim"]
        self.vae = VAE(z_dim=self.z_dim, nc=3,
                       latent_scale=_LATENT_SCALE[self.num_classes])
        self.discriminator = Discriminator(z_dim=self.z_dim)
        self.adversary_param = kwargs["vaal_adversary_param"]
        self.lr_discriminator = kwargs["lr_discriminator"]
        self.lr_vae = kwargs["lr_vae"]

    # ------------------------------------------------------------------ #
    # query: most-unlabeled-looking by discriminator score on VAE mu
    # ------------------------------------------------------------------ #

    @torch.no_grad()
    def query(self, budget):
        candidates = self.available_query_idxs()
        take = int(min(len(candidates), budget))
        for m in (self.net, self.vae, self.discriminator):
            m.eval()
        self.vae.to(self.device)
        self.discriminator.to(self.device)

        scores = torch.empty(len(candidates), device=self.device)
        loader = DataLoader(Subset(self.al_set, candidates), shuffle=False,
                            **self.train_args["loader_te_args"], drop_last=False)
        row = 0
        for x, _y, _idxs in loader:
            mu = self.vae(x.to(self.device))[3]
            d = self.discriminator(mu).view(-1)
            scores[row:row + d.numel()] = d
            row += d.numel()
        # low discriminator output = "looks unlabeled" -> query those first
        picked = torch.topk(scores.neg(), take).indices.cpu().numpy()
        return candidates[picked], take

    def init_network_weights(self):
        super().init_network_weights()
        self.vae.weight_init()
        self.discriminator.weight_init()

    # ------------------------------------------------------------------ #
    # co-training via the base trainer's hooks
    # ------------------------------------------------------------------ #

    def _setup_aux(self, device, rank):
        pool_idxs = self.available_query_idxs(boolean=False, shuffle=False)
        unl = Subset(self.train_set, pool_idxs)
        args = dict(self.train_args["loader_tr_args"])
        args["batch_size"] = max(1, int(args["batch_size"] / self.world_size))
        self._unl_sampler = None
        if self.world_size > 1:
            self._unl_sampler = torch.utils.data.distributed.DistributedSampler(
                unl, num_replicas=self.world_size, rank=rank, shuffle=True)
        self._loader_unl = DataLoader(
            unl, shuffle=self._unl_sampler is None, **args, drop_last=False,
            sampler=self._unl_sampler, pin_memory=device.type == "cuda")

        self.vae = self.vae.to(device).train()
        self.discriminator = self.discriminator.to(device).train()
        if self.world_size > 1:
            self.vae = BucketedDDP(self.vae)
            self.discriminator = BucketedDDP(self.discriminator)
        self._optim_vae = FusedAdam(self.vae.parameters(), lr=self.lr_vae)
        self._optim_disc = FusedAdam(self.discriminator.parameters(),
                                     lr=self.lr_discriminator)
        sched, sched_args = (self.train_args["lr_scheduler"],
                             self.train_args["lr_scheduler_args"])
        self._scheds_aux = [build_scheduler(sched, self._optim_vae, **sched_args),
                            build_scheduler(sched, self._optim_disc, **sched_args)]

    def _epoch_start(self, epoch):
        self.vae.train()
        self.discriminator.train()
        if self._unl_sampler is not None:
            self._unl_sampler.set_epoch(epoch)

    def _epoch_end(self):
        for s in self._scheds_aux:
            s.step()

    def _teardown_aux(self):
        if isinstance(self.vae, BucketedDDP):
            self.vae = self.vae.module
        if isinstance(self.discriminator, BucketedDDP):
            self.discriminator = self.discriminator.module
        self._loader_unl = self._unl_sampler = None
        self._optim_vae = self._optim_disc = self._scheds_aux = None

    # ------------------------------------------------------------------ #

    @staticmethod
    def _elbo(crop, recon, mu, logvar, beta=1.0):
        """Reconstruction + beta*KL for one slice (vaal_sampler.py:276-280)."""
        kld = -0.5 * torch.sum(1 + logvar - mu.pow(2) - logvar.exp())
        return F.mse_loss(recon, crop) + beta * kld

    def _step_aux_model(self, model, optimizer, loss):
        optimizer.zero_grad(set_to_none=True)
        loss.backward()
        if isinstance(model, BucketedDDP):
            model.finalize_grads()
        optimizer.step()

    def _train(self, rank, epoch, loader_tr, optimizer, criterion, step,
               graphed=None):
        """Co-training epoch. Per labeled batch: (1) classifier step (graph-
        replayed when the base trainer captured one), (2) VAE step on the
        concatenated labeled+unlabeled batch, (3) discriminator step on
        post-update mu (fresh no_grad forward, matching the reference's
        ordering semantics)."""
        unl_iter = iter(self._loader_unl)
        vae_core = self.vae.module if isinstance(self.vae, BucketedDDP) else self.vae
        bce = F.binary_cross_entropy
        for batch_idx, (x, y, _idxs) in enumerate(loader_tr):
            try:
                xu = next(unl_iter)[0]
            except StopIteration:
                unl_iter = iter(self._loader_unl)
                xu = next(unl_iter)[0]

            # (1) classifier
            if graphed is not None:
                loss = graphed.step(x, y)
                x = x.to(self.device, non_blocking=True)
            else:
                x = x.to(self.device, non_blocking=True)
                y = y.to(self.device, non_blocking=True)
                optimizer.zero_grad(set_to_none=True)
                loss = criterion(self.net(x), y)
                loss.backward()
                if isinstance(self.net, BucketedDDP):
                    self.net.finalize_grads()
                optimizer.step()
            self._log_train_batch(rank, epoch, batch_idx, loader_tr, loss)

            xu = xu.to(self.device, non_blocking=True)
            nl = x.size(0)
            both = torch.cat([x, xu])
            vae_core.set_crop_seed(int(np.random.randint(0, 10000)))

            # (2) VAE: one concatenated forward covers labeled + transductive
            crop, recon, _z, mu, logvar = self.vae(both)
            vae_loss = (self._elbo(crop[:nl], recon[:nl], mu[:nl], logvar[:nl])
                        + self._elbo(crop[nl:], recon[nl:], mu[nl:], logvar[nl:]))
            d_adv = self.discriminator(mu).view(-1)
            want_labeled = torch.ones_like(d_adv)
            adv = (bce(d_adv[:nl], want_labeled[:nl])
                   + bce(d_adv[nl:], want_labeled[nl:]))
            self._step_aux_model(self.vae, self._optim_vae,
                                 vae_loss + self.adversary_param * adv)

            # (3) discriminator on the updated VAE's mu
            with torch.no_grad():
                mu_post = self.vae(both)[3]
            d = self.discriminator(mu_post).view(-1)
            dsc_loss = (bce(d[:nl], torch.ones(nl, device=d.device))
                        + bce(d[nl:], torch.zeros(d.numel() - nl, device=d.device)))
            self._step_aux_model(self.discriminator, self._optim_disc, dsc_loss)
            step += 1
        return step
