"""Bucketed gradient all-reduce overlapped with backward — RCCL over xGMI.

SURVEY.md §2.13 RCCL integration layer: "bf16 grad buckets per DDP step;
7 p2p xGMI links × ≈153 GB/s per GPU ⇒ ring all-reduce is per-link bound —
choose bucket sizes for that". Buckets are contiguous ranges of the
FlatParamSpace grad buffer (no copies); each bucket's all-reduce launches on
a dedicated comm stream as soon as its last gradient lands (via
post-accumulate-grad hooks), overlapping communication with the rest of
backward. Default bucket ≈ 64 MiB: big enough to amortize RCCL launch and
ring pipelining over one xGMI link, small enough to overlap the tail.
"""
from __future__ import annotations

import os
from typing import List

import torch
import torch.distributed as dist

from .flat import FlatParamSpace, _aligned


class _Bucket:
    __slots__ = ("start", "end", "param_ids", "pending", "work", "ev")

    def __init__(self, start: int, end: int):
        self.start, self.end = start, end
        self.param_ids: List[int] = []
        self.pending = 0
        self.work = None
        self.ev = None  # (start_event, end_event) when comm timing is on


class BucketedDDP:
    def __init__(self, flat: FlatParamSpace, bucket_mb: float | None = None,
                 process_group=None):
        self.flat = flat
        self.pg = process_group
        self.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self.enabled = self.world > 1
        if bucket_mb is None:
            bucket_mb = float(os.environ.get("KF_DDP_BUCKET_MB", "64"))
        self.use_cuda = flat.device.type == "cuda"
        self.comm_stream = (torch.cuda.Stream(device=flat.device)
                           if self.use_cuda and self.enabled else None)
        self._avg_supported = (self.enabled and dist.is_initialized()
                               and dist.get_backend(process_group) == "nccl")
        self._sync = False
        # comm/compute overlap stats for the scaling bench (KF_COMM_STATS):
        # CUDA events bracket each bucket's all-reduce on the comm stream;
        # last_comm_ms is their summed GPU time for the latest step.
        self._timing = (os.environ.get("KF_COMM_STATS") == "1"
                        and self.comm_stream is not None)
        self.last_comm_ms: float | None = None
        self.bucket_mb = bucket_mb
        self._build_buckets(int(bucket_mb * 1024 * 1024 / flat.grad.element_size()))
        if self.enabled:
            self._register_hooks()
            # broadcast initial parameters so every rank starts identical
            # (src = the group's first member — a global rank, which is not
            # 0 for non-leading DP groups of a TP x DP mesh)
            src = (dist.get_global_rank(self.pg, 0)
                   if self.pg is not None else 0)
            dist.broadcast(flat.data, src=src, group=self.pg)

    def _build_buckets(self, bucket_numel: int):
        self.buckets: List[_Bucket] = []
        self.param_bucket = {}
        cur = _Bucket(0, 0)
        for i, (off, n) in enumerate(self.flat.slices):
            end = off + _aligned(n)
            if cur.param_ids and end - cur.start > bucket_numel:
                cur.end = off
                self.buckets.append(cur)
                cur = _Bucket(off, off)
            cur.param_ids.append(i)
            self.param_bucket[i] = cur
            cur.end = end
        if cur.param_ids:
            self.buckets.append(cur)

    def _register_hooks(self):
        for i, p in enumerate(self.flat.params):
            bucket = self.param_bucket[i]

            def hook(param, bucket=bucket):
                if not self._sync:  # grad-accum micro-steps skip comm
                    return
                bucket.pending -= 1
                if bucket.pending == 0:
                    self._launch(bucket)

            p.register_post_accumulate_grad_hook(hook)

    def _launch(self, bucket: _Bucket):
        view = self.flat.grad[bucket.start:bucket.end]
        op = dist.ReduceOp.AVG if self._avg_supported else dist.ReduceOp.SUM
        if self.comm_stream is not None:
            self.comm_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self.comm_stream):
                if self._timing:
                    e0 = torch.cuda.Event(enable_timing=True)
                    e1 = torch.cuda.Event(enable_timing=True)
                    e0.record(self.comm_stream)
                    bucket.work = dist.all_reduce(view, op=op, group=self.pg,
                                                  async_op=True)
                    e1.record(self.comm_stream)
                    bucket.ev = (e0, e1)
                else:
                    bucket.work = dist.all_reduce(view, op=op, group=self.pg,
                                                  async_op=True)
        else:
            bucket.work = dist.all_reduce(view, op=op, group=self.pg,
                                          async_op=True)

    def prepare_step(self):
        """Arm bucket counters before each backward."""
        if not self.enabled:
            return
        self._sync = True
        for b in self.buckets:
            b.pending = len(b.param_ids)
            b.work = None

    def finalize(self):
        """Wait for all in-flight all-reduces; call after backward, before
        the optimizer step."""
        if not self.enabled:
            return
        for b in self.buckets:
            if b.pending != 0 and b.work is None:
                # grads some params never produced (unused param): reduce now
                self._launch(b)
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
        if self.comm_stream is not None:
            torch.cuda.current_stream().wait_stream(self.comm_stream)
        if self._timing:
            torch.cuda.synchronize()
            ms = 0.0
            for b in self.buckets:
                if b.ev is not None:
                    ms += b.ev[0].elapsed_time(b.ev[1])
                    b.ev = None
            self.last_comm_ms = ms
        self._sync = False
        if not self._avg_supported:
            self.flat.grad.div_(self.world)
