"""Bucketed-gradient data parallelism over RCCL/xGMI.

MI355X scaling model: one process per GPU, gradients
all-reduced over RCCL (`torch.distributed` backend "nccl" IS RCCL on ROCm;
"gloo" for CPU tests).  xGMI is point-to-point (7 links x ~153 GB/s per GPU),
so a ring all-reduce is per-link bound; the R50 gradient payload is ~51 MB in
bf16 — we bucket at ~25 MB and launch each bucket's all-reduce as soon as its
last gradient lands in backward, overlapping communication with the rest of
backward.

Statistics under DP are per-rank by default (stock-DDP-like BN semantics —
the paper's stats are per-minibatch); `sync_stats()` optionally all-reduces
the EMA buffers (a tiny collective: G*g*g + C floats per site), and
`broadcast_buffers()` aligns eval stats across ranks.
"""
from __future__ import annotations

import os
from typing import List, Optional

import torch
import torch.distributed as dist


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1


def init_distributed_from_env(backend: Optional[str] = None):
    """torchrun-style init: reads RANK/WORLD_SIZE/LOCAL_RANK/MASTER_*."""
    if "RANK" not in os.environ or int(os.environ.get("WORLD_SIZE", "1")) <= 1:
        return 0, 1, 0
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    world = dist.get_world_size()
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return rank, world, local_rank


class _Bucket:
    def __init__(self, params: List[torch.nn.Parameter], dtype=None):
        self.params = params
        self.numel = sum(p.numel() for p in params)
        self.dtype = dtype  # None -> param dtype (grad-as-view fast path)
        self.buffer: Optional[torch.Tensor] = None
        self.views = {}
        self.offsets = {}
        self.pending = 0
        self.work = None

    def ensure_buffer(self):
        if self.buffer is None:
            p0 = self.params[0]
            dt = self.dtype or p0.dtype
            self.buffer = torch.zeros(self.numel, dtype=dt, device=p0.device)
            off = 0
            for p in self.params:
                n = p.numel()
                self.offsets[p] = off
                if self.dtype is None:
                    self.views[p] = self.buffer[off:off + n].view(p.shape)
                off += n


class BucketedDataParallel:
    """Gradient synchronization engine (not an nn.Module wrapper — the model
    stays untouched; call ``sync()`` between backward() and optimizer.step()).

    Buckets are assembled in reverse parameter order (approximate backward
    completion order); each bucket's async all-reduce launches from a
    post-accumulate-grad hook as soon as its last grad is ready.

    Grad-as-bucket-view: when the bucket dtype matches the param dtype
    (default), ``p.grad`` is redirected to a view of the flat bucket buffer
    the first time it lands, so later steps accumulate straight into the
    buffer and the reduce needs NO copy-in/copy-out (FusedSGD/FusedAdam keep
    grad storage stable by zeroing grads in-kernel).  Gradients are
    pre-divided by world size before the SUM all-reduce so bf16 sums at
    world size 8 cannot overflow and carry less accumulated rounding than
    sum-then-divide.  ``reduce_dtype=torch.float32`` instead keeps an fp32
    bucket (copy-in/copy-out each step) for exact accumulation with bf16
    params.
    """

    def __init__(self, model: torch.nn.Module, bucket_cap_mb: float = 25.0,
                 process_group=None, average: bool = True,
                 reduce_dtype: Optional[torch.dtype] = None):
        self.model = model
        self.group = process_group
        self.average = average
        self.world = dist.get_world_size(process_group) if is_distributed() else 1
        self.params = [p for p in model.parameters() if p.requires_grad]
        self.enabled = self.world > 1
        self.reduce_dtype = reduce_dtype

        self.buckets: List[_Bucket] = []
        self._param_bucket = {}
        # while True, grad-ready hooks do NOT launch reductions (gradient
        # accumulation across several backwards; use accumulate()):
        self.accumulating = False
        if self.enabled:
            self._build_buckets(bucket_cap_mb)
            self._install_hooks()
            self.broadcast_parameters()

    # -- setup ---------------------------------------------------------------
    def _bucket_dtype_for(self, p: torch.nn.Parameter):
        if self.reduce_dtype is not None and self.reduce_dtype != p.dtype:
            return self.reduce_dtype
        return None  # param dtype -> grad-as-view

    def _build_buckets(self, cap_mb: float):
        cap = int(cap_mb * 1024 * 1024)
        cur, cur_bytes, cur_key = [], 0, None
        for p in reversed(self.params):
            key = (p.dtype, p.device)
            if cur and key != cur_key:
                self.buckets.append(_Bucket(cur, self._bucket_dtype_for(cur[0])))
                cur, cur_bytes = [], 0
            cur_key = key
            cur.append(p)
            cur_bytes += p.numel() * p.element_size()
            if cur_bytes >= cap:
                self.buckets.append(_Bucket(cur, self._bucket_dtype_for(p)))
                cur, cur_bytes = [], 0
        if cur:
            self.buckets.append(_Bucket(cur, self._bucket_dtype_for(cur[0])))
        for b in self.buckets:
            for p in b.params:
                self._param_bucket[p] = b

    def _install_hooks(self):
        for p in self.params:
            p.register_post_accumulate_grad_hook(self._on_grad)

    # -- runtime -------------------------------------------------------------
    def _on_grad(self, p: torch.nn.Parameter):
        b = self._param_bucket.get(p)
        if b is None:
            return
        b.ensure_buffer()
        if b.dtype is None:
            v = b.views[p]
            if p.grad is not v:
                # first step (or after zero_grad(set_to_none=True)): move the
                # freshly accumulated grad into the bucket and alias it
                v.copy_(p.grad.detach().reshape(p.shape))
                p.grad = v
        if self.accumulating:
            # a reduction launched now would race with the NEXT backward's
            # accumulation into the same buffer — defer to the final one
            return
        b.pending += 1
        if b.pending >= len(b.params):
            self._launch(b)

    def accumulate(self):
        """Context manager: suppress hook-launched reductions for backwards
        whose gradients should accumulate locally (call ``sync()`` after the
        final backward as usual)."""
        import contextlib

        @contextlib.contextmanager
        def _ctx():
            prev = self.accumulating
            self.accumulating = True
            try:
                yield
            finally:
                self.accumulating = prev

        return _ctx()

    def _launch(self, b: _Bucket):
        if b.work is not None:
            return
        b.ensure_buffer()
        if b.dtype is not None:
            # fp32-reduce path: copy grads in (upcast)
            for p in b.params:
                off = b.offsets[p]
                n = p.numel()
                if p.grad is not None:
                    b.buffer[off:off + n].copy_(p.grad.detach().reshape(-1))
                else:
                    b.buffer[off:off + n].zero_()
        else:
            # grad-as-view path: zero the segments of params with no grad
            for p in b.params:
                if p.grad is not b.views[p]:
                    if p.grad is None:
                        b.views[p].zero_()
                    else:
                        b.views[p].copy_(p.grad.detach().reshape(p.shape))
                        p.grad = b.views[p]
        if self.average:
            b.buffer.div_(self.world)
        b.work = dist.all_reduce(b.buffer, op=dist.ReduceOp.SUM,
                                 group=self.group, async_op=True)

    def sync(self):
        """Wait for all bucket reductions; grads end up averaged in place."""
        if not self.enabled:
            return
        for b in self.buckets:
            if b.work is None:
                self._launch(b)
        for b in self.buckets:
            b.work.wait()
            if b.dtype is not None:
                for p in b.params:
                    off = b.offsets[p]
                    n = p.numel()
                    g = b.buffer[off:off + n].reshape(p.shape).to(p.dtype)
                    if p.grad is None:
                        p.grad = g.clone()
                    else:
                        p.grad.copy_(g)
            else:
                for p in b.params:
                    if p.grad is None:
                        p.grad = b.views[p]
            b.work = None
            b.pending = 0

    # -- parameter / stats collectives --------------------------------------
    def broadcast_parameters(self, src: int = 0):
        if not self.enabled:
            return
        with torch.no_grad():
            for t in self.model.state_dict().values():
                if torch.is_tensor(t) and t.numel() > 0 and not t.is_sparse:
                    dist.broadcast(t, src=src, group=self.group)

    def broadcast_buffers(self, src: int = 0):
        if not self.enabled:
            return
        with torch.no_grad():
            for b in self.model.buffers():
                dist.broadcast(b, src=src, group=self.group)

    def sync_stats(self):
        """Average floating-point EMA buffers across ranks ('sync' stats mode:
        the per-site covariances/means are tiny collectives)."""
        if not self.enabled:
            return
        with torch.no_grad():
            for b in self.model.buffers():
                if b.dtype.is_floating_point:
                    dist.all_reduce(b, op=dist.ReduceOp.SUM, group=self.group)
                    b.div_(self.world)
