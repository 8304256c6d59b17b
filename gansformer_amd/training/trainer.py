"""GANTrainer: one full GANsformer training step (used by both the
training loop and bench.py, so benchmarks time exactly the production
step: D main, lazy R1, G main, lazy path-length, EMA, optimizer steps,
bucketed RCCL all-reduce).
"""

from __future__ import annotations

import torch

from ..parallel import GradReducer
from .loss import (PathLengthRegularizer, d_logistic_loss,
                   g_nonsaturating_loss, r1_penalty)
from .loop_utils import ema_update, lazy_adam


class GANTrainer:
    def __init__(self, G, D, Gs, device, batch_gpu, batch_size,
                 g_lr=0.002, d_lr=0.002, beta1=0.0, beta2=0.99, eps=1e-8,
                 gamma=10.0, d_reg_interval=16, g_reg_interval=8,
                 pl_weight=2.0, pl_decay=0.01, ema_kimg=10.0,
                 ema_rampup=0.05, rounds=1, bucket_mb=25.0, r1_fp32=False,
                 label_fn=None):
        self.G, self.D, self.Gs = G, D, Gs
        self.device = device
        self.batch_gpu = batch_gpu
        self.batch_size = batch_size
        self.rounds = rounds
        self.gamma = gamma
        self.r1_fp32 = r1_fp32
        self.d_reg_interval = d_reg_interval
        self.g_reg_interval = g_reg_interval
        self.ema_kimg = ema_kimg
        self.ema_rampup = ema_rampup
        self.pl_reg = PathLengthRegularizer(decay=pl_decay, weight=pl_weight)
        self.g_opt = lazy_adam(G.parameters(), g_lr, (beta1, beta2), eps,
                               g_reg_interval)
        self.d_opt = lazy_adam(D.parameters(), d_lr, (beta1, beta2), eps,
                               d_reg_interval)
        self.g_red = GradReducer(G, bucket_mb=bucket_mb)
        self.d_red = GradReducer(D, bucket_mb=bucket_mb)
        # class-conditional training: label_fn(n) -> [n, label_dim]
        # samples generator labels from the dataset's label marginal
        self.conditional = getattr(G, "label_dim", 0) > 0
        self.label_fn = label_fn
        if self.conditional and label_fn is None:
            raise ValueError("conditional G (label_dim > 0) needs label_fn")

    def step(self, next_real_fn, step_idx, cur_nimg, start_nimg=0,
             logger=None):
        """Run one full training step. next_real_fn() -> [B,C,H,W] in [-1,1]."""
        G, D = self.G, self.D
        rounds = self.rounds
        do_d_reg = self.d_reg_interval > 0 and step_idx % self.d_reg_interval == 0
        do_g_reg = self.g_reg_interval > 0 and step_idx % self.g_reg_interval == 0
        rep = logger.report if logger is not None else (lambda *a: None)
        cond = self.conditional

        def unpack(r):
            return r if isinstance(r, (tuple, list)) else (r, None)

        # ---------------- D main ----------------
        self.d_opt.zero_grad(set_to_none=True)
        self.d_red.prepare(rounds)
        for _ in range(rounds):
            real, rlab = unpack(next_real_fn())
            z = G.sample_z(self.batch_gpu, device=self.device)
            flab = self.label_fn(self.batch_gpu) if cond else None
            with torch.no_grad():
                fake = G(z, label=flab, style_mixing=True)
            real_logits = D(real, label=rlab) if cond else D(real)
            fake_logits = D(fake, label=flab) if cond else D(fake)
            loss_d = d_logistic_loss(real_logits, fake_logits) / rounds
            loss_d.backward()
            rep("Loss/D", loss_d.detach() * rounds)
            rep("Loss/scores_real", real_logits.detach().mean())
            rep("Loss/scores_fake", fake_logits.detach().mean())
        self.d_red.finalize()
        self.d_opt.step()

        # ---------------- D reg (lazy R1) ----------------
        if do_d_reg:
            self.d_opt.zero_grad(set_to_none=True)
            self.d_red.prepare(rounds)
            for _ in range(rounds):
                real, rlab = unpack(next_real_fn())
                real = real.requires_grad_(True)
                real_logits = D(real, label=rlab, force_fp32=self.r1_fp32) \
                    if cond else D(real, force_fp32=self.r1_fp32)
                r1 = r1_penalty(real_logits, real)
                ((self.gamma * r1 * self.d_reg_interval) / rounds).backward()
                rep("Loss/r1", r1.detach())
            self.d_red.finalize()
            self.d_opt.step()

        # ---------------- G main ----------------
        self.g_opt.zero_grad(set_to_none=True)
        self.g_red.prepare(rounds)
        for _ in range(rounds):
            z = G.sample_z(self.batch_gpu, device=self.device)
            flab = self.label_fn(self.batch_gpu) if cond else None
            fake = G(z, label=flab, style_mixing=True, update_w_avg=True)
            fake_logits = D(fake, label=flab) if cond else D(fake)
            loss_g = g_nonsaturating_loss(fake_logits) / rounds
            loss_g.backward()
            rep("Loss/G", loss_g.detach() * rounds)
        self.g_red.finalize()
        self.g_opt.step()

        # ---------------- G reg (lazy path length) ----------------
        if do_g_reg:
            self.g_opt.zero_grad(set_to_none=True)
            self.g_red.prepare(rounds)
            pl_batch = max(self.batch_gpu // 2, 1)
            for _ in range(rounds):
                z = G.sample_z(pl_batch, device=self.device)
                flab = self.label_fn(pl_batch) if cond else None
                ws = G.mapping(z, label=flab)
                fake = G.synthesis(ws)
                pl = self.pl_reg(fake, ws)
                ((pl * self.g_reg_interval) / rounds).backward()
                rep("Loss/pl", pl.detach())
            self.g_red.finalize()
            self.g_opt.step()

        # ---------------- EMA ----------------
        ema_nimg = self.ema_kimg * 1000
        if self.ema_rampup is not None:
            ema_nimg = min(ema_nimg,
                           max(cur_nimg - start_nimg, 1) * self.ema_rampup)
        beta = 0.5 ** (self.batch_size / max(ema_nimg, 1e-8))
        ema_update(self.Gs, G, beta)

    @torch.no_grad()
    def sync_ranks(self):
        """Reconcile per-rank-divergent training state across DP ranks.

        Parameters stay in lockstep through the grad all-reduce, but
        `MappingNetwork.w_avg` and `PathLengthRegularizer.pl_mean` update
        from rank-local batches and drift apart; `Gs` inherits the drift
        via its buffers. All-reduce them to the cross-rank mean so that a
        snapshot saved by rank 0 matches every rank's live state (called
        at snapshot/metric cadence by the training loop).
        """
        import torch.distributed as dist
        if not dist.is_initialized() or dist.get_world_size() == 1:
            return
        inv = 1.0 / dist.get_world_size()
        bufs = []
        wa = getattr(getattr(self.G, "mapping", None), "w_avg", None)
        if wa is not None:
            bufs.append(wa)
        if self.pl_reg.pl_mean is not None:
            bufs.append(self.pl_reg.pl_mean)
        for t in bufs:
            dist.all_reduce(t, op=dist.ReduceOp.SUM)
            t.mul_(inv)
        # propagate the reconciled w_avg into the EMA copy
        gwa = getattr(getattr(self.Gs, "mapping", None), "w_avg", None)
        if wa is not None and gwa is not None:
            gwa.copy_(wa)

    def state_dict(self):
        return dict(g_opt=self.g_opt.state_dict(),
                    d_opt=self.d_opt.state_dict(),
                    pl_reg=self.pl_reg.state_dict())

    def load_state_dict(self, state):
        if "g_opt" in state:
            self.g_opt.load_state_dict(state["g_opt"])
        if "d_opt" in state:
            self.d_opt.load_state_dict(state["d_opt"])
        if "pl_reg" in state:
            self.pl_reg.load_state_dict(state["pl_reg"])
