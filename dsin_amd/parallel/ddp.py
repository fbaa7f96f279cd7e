"""Data parallelism over RCCL/xGMI: bucketed gradient all-reduce overlapped
with backward.

The reference is single-process single-GPU (zero collective call sites —
SURVEY.md section 2c); DP is the one parallelism strategy this rebuild adds.
Design sized for the MI355X node fabric: xGMI is point-to-point (7 links x
~153 GB/s per GPU) and the whole gradient set is only ~40 MB fp32, so the
all-reduce is latency-, not bandwidth-, dominated — few (4-8) MB-scale
buckets, each launched as soon as its last gradient materializes in
backward, overlap the ring latency with remaining backward compute.

Implementation: flat preallocated bucket buffers in reverse parameter order
(backward produces gradients roughly in reverse), per-parameter
post-accumulate-grad hooks copy into the flat buffer and launch an async
``all_reduce`` on the bucket when it completes; ``finalize()`` waits on all
works, scales by 1/world and copies back. Optional bf16 communication dtype
halves bytes on the wire (fp32 master grads are kept). Works over both the
"nccl" (== RCCL on ROCm) and "gloo" (CPU tests) backends.
"""

from __future__ import annotations

import os
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def init_distributed(backend: Optional[str] = None) -> int:
    """Initialize torch.distributed from torchrun-style env vars.
    Returns local rank. No-op (returns 0) when WORLD_SIZE is absent or 1."""
    ws = int(os.environ.get("WORLD_SIZE", "1"))
    if ws <= 1:
        return 0
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend)
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return local_rank


class _Bucket:
    __slots__ = ("params", "offsets", "flat", "pending", "work", "launched")

    def __init__(self, params: List[torch.nn.Parameter], device, dtype):
        self.params = params
        self.offsets = []
        off = 0
        for p in params:
            self.offsets.append(off)
            off += p.numel()
        self.flat = torch.zeros(off, device=device, dtype=dtype)
        self.pending = 0
        self.work = None
        self.launched = False

    def reset(self):
        self.pending = len(self.params)
        self.work = None
        self.launched = False


class GradReducer:
    """Bucketed overlapped gradient all-reduce.

    Usage per iteration:
        reducer.prepare()      # before backward
        loss.backward()        # hooks fire, buckets launch as they fill
        reducer.finalize()     # wait + average into param.grad
    """

    def __init__(self, params: Sequence[torch.nn.Parameter],
                 bucket_bytes: int = 8 * 1024 * 1024,
                 comm_dtype: Optional[torch.dtype] = None):
        self.active = is_distributed() and world_size() > 1
        self.params = [p for p in params if p.requires_grad]
        if not self.params:
            self.active = False
        self.comm_dtype = comm_dtype
        self._hooks = []
        self._buckets: List[_Bucket] = []
        self._param_bucket = {}
        if not self.active:
            return
        device = self.params[0].device
        dtype = comm_dtype or self.params[0].dtype
        elt = torch.tensor([], dtype=dtype).element_size()
        cap = max(bucket_bytes // elt, 1)
        # reverse order: backward computes gradients tail-first
        cur: List[torch.nn.Parameter] = []
        size = 0
        for p in reversed(self.params):
            cur.append(p)
            size += p.numel()
            if size >= cap:
                self._buckets.append(_Bucket(cur, device, dtype))
                cur, size = [], 0
        if cur:
            self._buckets.append(_Bucket(cur, device, dtype))
        for b in self._buckets:
            for i, p in enumerate(b.params):
                self._param_bucket[id(p)] = (b, i)
                self._hooks.append(
                    p.register_post_accumulate_grad_hook(self._make_hook(b, i)))

    def _make_hook(self, bucket: _Bucket, idx: int):
        def hook(param: torch.nn.Parameter):
            if bucket.pending == 0:
                return  # prepare() not called (e.g. eval backward) — ignore
            off = bucket.offsets[idx]
            n = param.numel()
            bucket.flat[off:off + n].copy_(param.grad.detach().reshape(-1))
            bucket.pending -= 1
            if bucket.pending == 0:
                self._launch(bucket)
        return hook

    def _launch(self, bucket: _Bucket):
        bucket.work = dist.all_reduce(bucket.flat, op=dist.ReduceOp.SUM, async_op=True)
        bucket.launched = True

    def prepare(self):
        if not self.active:
            return
        for b in self._buckets:
            b.reset()

    def finalize(self):
        if not self.active:
            return
        ws = float(world_size())
        for b in self._buckets:
            if not b.launched:
                # some params produced no grad this step: zero their slices
                for i, p in enumerate(b.params):
                    if p.grad is None:
                        off = b.offsets[i]
                        b.flat[off:off + p.numel()].zero_()
                    elif b.pending > 0:
                        off = b.offsets[i]
                        b.flat[off:off + p.numel()].copy_(p.grad.detach().reshape(-1))
                self._launch(b)
        for b in self._buckets:
            if b.work is not None:
                b.work.wait()
            for i, p in enumerate(b.params):
                off = b.offsets[i]
                avg = b.flat[off:off + p.numel()].div(ws)
                if p.grad is None:
                    p.grad = avg.to(p.dtype).view_as(p).clone()
                else:
                    p.grad.copy_(avg.view_as(p))

    def broadcast_params(self):
        """Rank-0 parameter broadcast at startup so replicas start identical."""
        if not self.active:
            return
        for p in self.params:
            dist.broadcast(p.data, src=0)

    def remove(self):
        for h in self._hooks:
            h.remove()


class _FlatBucket:
    __slots__ = ("params", "lo", "hi", "pending", "work", "launched")

    def __init__(self, entries, lo: int, hi: int):
        self.params = entries          # [(param, ofs, numel), ...]
        self.lo = lo
        self.hi = hi
        self.pending = 0
        self.work = None
        self.launched = False


class FlatGradReducer:
    """Overlapped bucketed all-reduce over a FusedAdam's flat gradient
    buffer (the fused-optimizer DDP path).

    The optimizer's flat_g is the communication buffer: parameters occupy
    contiguous forward-order slices, so a run of consecutive parameters in
    REVERSE order (the order backward produces gradients) covers one
    contiguous tail range of flat_g. Each bucket's last-arriving gradient
    (post-accumulate-grad hook) triggers ONE _foreach_copy_ of the bucket's
    stolen .grad tensors into flat_g followed by an async all_reduce on the
    contiguous range — the xGMI transfer overlaps the rest of backward
    instead of serializing after it (SURVEY 2c sizing:
    ~20-40 MB of grads over 7x153 GB/s links is latency-dominated, so 4-8 MB
    buckets, few launches). finalize() flushes buckets whose parameters
    produced no gradient, waits on all works (stream-level wait on RCCL)
    and scales by 1/world_size.
    """

    def __init__(self, opt, bucket_bytes: int = 8 * 1024 * 1024):
        self.opt = opt
        self.active = is_distributed() and world_size() > 1
        self._hooks = []
        self._buckets: List[_FlatBucket] = []
        if not self.active:
            return
        elt = opt.flat_g.element_size()
        cap = max(bucket_bytes // elt, 1)
        cur, size = [], 0
        for p, (ofs, n) in reversed(list(zip(opt.params, opt._slices))):
            cur.append((p, ofs, n))
            size += n
            if size >= cap:
                self._add_bucket(cur)
                cur, size = [], 0
        if cur:
            self._add_bucket(cur)
        for b in self._buckets:
            for p, _, _ in b.params:
                self._hooks.append(
                    p.register_post_accumulate_grad_hook(self._make_hook(b)))

    def _add_bucket(self, entries):
        lo = min(ofs for _, ofs, _ in entries)
        hi = max(ofs + n for _, ofs, n in entries)
        self._buckets.append(_FlatBucket(list(entries), lo, hi))

    def _make_hook(self, bucket: _FlatBucket):
        def hook(_param):
            if bucket.pending == 0:
                return  # prepare() not called (e.g. eval backward) — ignore
            bucket.pending -= 1
            if bucket.pending == 0:
                self._copy_launch(bucket)
        return hook

    def _copy_launch(self, bucket: _FlatBucket):
        g = self.opt.flat_g
        dsts, srcs = [], []
        for p, ofs, n in bucket.params:
            if p.grad is None:
                g[ofs:ofs + n].zero_()
            else:
                dsts.append(g[ofs:ofs + n])
                srcs.append(p.grad.reshape(-1))
        if dsts:
            torch._foreach_copy_(dsts, srcs)
        bucket.work = dist.all_reduce(g[bucket.lo:bucket.hi],
                                      op=dist.ReduceOp.SUM, async_op=True)
        bucket.launched = True

    def prepare(self):
        if not self.active:
            return
        for b in self._buckets:
            b.pending = len(b.params)
            b.work = None
            b.launched = False

    def finalize(self):
        """Flush + wait + average. After this, opt.flat_g holds the mean
        gradient and opt.gather_grads() must NOT be called (the hooks
        already gathered)."""
        if not self.active:
            return
        for b in self._buckets:
            if not b.launched:
                self._copy_launch(b)
        for b in self._buckets:
            if b.work is not None:
                b.work.wait()
            b.pending = 0
        self.opt.flat_g.div_(float(world_size()))

    def remove(self):
        for h in self._hooks:
            h.remove()
