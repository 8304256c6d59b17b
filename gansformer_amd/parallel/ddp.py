"""Bucketed gradient all-reduce, overlapped with backward.

Replaces the reference's per-variable `nccl_ops.all_sum` at
`Optimizer.apply_updates` (ref src/dnnlib/tflib/optimizer.py [R]) with an
MI355X-appropriate scheme: grads are packed into ~bucket_mb flat buffers
and all-reduced asynchronously as soon as every grad in a bucket is
accumulated (post-accumulate-grad hooks), overlapping communication with
the rest of backward. xGMI collectives are per-link bound, so few large
buckets beat many small ones; grad volume here is ~30-120 MB, i.e. a
handful of buckets.

G and D get separate GradReducers (they step alternately). Lazy-reg
steps that populate only a subset of grads are handled by finalize():
buckets that never became complete are flushed there (missing grads
contribute zeros on every rank alike, so averages stay correct).
"""

from __future__ import annotations

import torch
import torch.distributed as dist


def broadcast_params(module: torch.nn.Module, src: int = 0):
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return
    for t in list(module.parameters()) + list(module.buffers()):
        dist.broadcast(t.detach(), src=src)


class _Bucket:
    __slots__ = ("params", "numel", "buffer", "work", "counts", "launched")

    def __init__(self):
        self.params = []
        self.numel = 0
        self.buffer = None
        self.work = None
        self.counts = {}
        self.launched = False


class GradReducer:
    def __init__(self, module: torch.nn.Module, bucket_mb: float = 25.0,
                 process_group=None, force_enabled: bool = False):
        """force_enabled drives the bucket/collective path even at
        world_size 1 — used by the single-GPU RCCL smoke test (gpurun
        boxes have one GPU and RCCL forbids 2 ranks per device)."""
        self.module = module
        self.group = process_group
        self.enabled = (dist.is_initialized()
                        and (dist.get_world_size() > 1 or force_enabled))
        self.world_size = dist.get_world_size() if self.enabled else 1
        self.buckets: list[_Bucket] = []
        self.param_bucket = {}
        if not self.enabled:
            return
        cap = int(bucket_mb * 1024 * 1024)
        # bucket in reverse parameter order ~ backward completion order
        params = [p for p in module.parameters() if p.requires_grad][::-1]
        b = _Bucket()
        for p in params:
            if b.params and (b.numel + p.numel()) * 4 > cap:
                self.buckets.append(b)
                b = _Bucket()
            b.params.append(p)
            b.numel += p.numel()
            self.param_bucket[p] = b
        if b.params:
            self.buckets.append(b)
        for p in params:
            p.register_post_accumulate_grad_hook(self._hook)

    def _hook(self, p):
        if not self.active:
            return
        b = self.param_bucket[p]
        # With gradient accumulation (rounds > 1), every backward pass
        # fires this hook once per participating param. Launching the
        # all-reduce before the LAST accumulation pass would ship partial
        # gradients and then overwrite the fully-accumulated p.grad in
        # finalize(). So a param only counts as ready once it has fired
        # `rounds` times; params absent from some round's graph simply
        # never complete the bucket and are flushed in finalize(), where
        # p.grad holds the full accumulated value.
        b.counts[p] = b.counts.get(p, 0) + 1
        if (not b.launched
                and all(b.counts.get(q, 0) >= self.rounds for q in b.params)):
            self._launch(b)

    def _launch(self, b: _Bucket):
        if b.launched:
            return
        b.launched = True
        dev = b.params[0].device
        if b.buffer is None or b.buffer.device != dev:
            b.buffer = torch.zeros(b.numel, dtype=torch.float32, device=dev)
        off = 0
        for p in b.params:
            n = p.numel()
            if p.grad is not None:
                b.buffer[off:off + n].copy_(p.grad.detach().reshape(-1))
            else:
                b.buffer[off:off + n].zero_()
            off += n
        b.work = dist.all_reduce(b.buffer, op=dist.ReduceOp.SUM,
                                 group=self.group, async_op=True)

    def prepare(self, rounds: int = 1):
        """Call before backward on a step whose grads should be reduced.

        `rounds` = number of gradient-accumulation backward passes that
        will run before finalize(); buckets launch eagerly only on the
        final pass (see _hook).
        """
        self.active = True
        self.rounds = max(int(rounds), 1)
        for b in self.buckets:
            b.counts = {}
            b.launched = False
            b.work = None

    active = False
    rounds = 1

    def finalize(self):
        """Flush partial buckets, wait for comms, write averaged grads back."""
        if not self.enabled or not self.active:
            self.active = False
            return
        for b in self.buckets:
            if not b.launched:
                self._launch(b)
        inv = 1.0 / self.world_size
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
            off = 0
            for p in b.params:
                n = p.numel()
                if p.grad is not None:
                    p.grad.detach().view(-1).copy_(b.buffer[off:off + n]).mul_(inv)
                off += n
        self.active = False
