"""Bucketed gradient all-reduce overlapped with backward (collective C1).

Hand-rolled (NOT torch.nn.parallel.DistributedDataParallel): gradients live
in the FlatParamSpace's flat buffer, buckets are contiguous ranges of that
buffer, and each bucket's RCCL all-reduce launches asynchronously the moment
its last gradient is accumulated — so communication rides under the rest of
backward.

xGMI sizing: each MI355X has 7 point-to-point xGMI links at
~153 GB/s; RCCL multi-ring spreads a big all-reduce over all 7 links
(~1 TB/s aggregate per GPU). Buckets default to 50 MB — large enough to
amortize ring latency on per-link-bound rings, small enough to start
reducing early in backward.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from .dist import get_world_size, is_distributed
from .flat import FlatParamSpace


class _Bucket:
    __slots__ = ("start", "end", "params", "pending", "work")

    def __init__(self, start: int, end: int, params: List[torch.nn.Parameter]):
        self.start = start
        self.end = end
        self.params = params
        self.pending = 0
        self.work = None


class DataParallelGrads:
    def __init__(
        self,
        space: FlatParamSpace,
        bucket_mb: int = 50,
        process_group: Optional[dist.ProcessGroup] = None,
    ):
        self.space = space
        self.pg = process_group
        self.enabled = is_distributed()
        self.require_reduce = True
        self.buckets: List[_Bucket] = []
        self._param_bucket = {}

        if not self.enabled:
            return

        bucket_bytes = bucket_mb * 1024 * 1024
        elem_size = space.flat_grad.element_size()
        # Build buckets walking segments in REVERSE model order: backward
        # produces late-layer grads first, so reverse buckets fill (and launch)
        # earliest. Each bucket is a contiguous flat range.
        segs = list(space.segments)
        i = len(segs) - 1
        while i >= 0:
            end_off = segs[i][1] + segs[i][2]
            j = i
            size = 0
            params: List[torch.nn.Parameter] = []
            while j >= 0 and (size == 0 or size + segs[j][2] * elem_size <= bucket_bytes):
                size += segs[j][2] * elem_size
                params.append(space.name_to_param[segs[j][0]])
                j -= 1
            start_off = segs[j + 1][1]
            bucket = _Bucket(start_off, end_off, params)
            self.buckets.append(bucket)
            for p in params:
                self._param_bucket[id(p)] = bucket
            i = j
        self._reset_pending()

        for p in space.params:
            p.register_post_accumulate_grad_hook(self._hook)

    # -- per-step lifecycle -----------------------------------------------
    def _reset_pending(self) -> None:
        for b in self.buckets:
            b.pending = len(b.params)
            b.work = None

    def _hook(self, p: torch.nn.Parameter) -> None:
        if not self.enabled or not self.require_reduce:
            return
        b = self._param_bucket.get(id(p))
        if b is None:
            return
        b.pending -= 1
        if b.pending == 0:
            self._launch(b)

    def _launch(self, b: _Bucket) -> None:
        view = self.space.flat_grad[b.start : b.end]
        b.work = dist.all_reduce(view, op=dist.ReduceOp.SUM, group=self.pg, async_op=True)

    def finalize(self) -> None:
        """Wait for all bucket reductions; average. Call after backward
        (last micro-step only when using gradient accumulation)."""
        if not self.enabled:
            return
        for b in self.buckets:
            if b.work is None and b.pending > 0:
                # params whose grads never materialized this step (unused):
                # reduce anyway so ranks stay consistent.
                self._launch(b)
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
        self.space.flat_grad.div_(
            dist.get_world_size(self.pg) if self.pg is not None else get_world_size()
        )
        self._reset_pending()
