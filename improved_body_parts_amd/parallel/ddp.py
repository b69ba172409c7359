"""Data-parallel gradient reduction: bucketed all-reduce overlapped with backward.

Replaces the reference's Apex ``DistributedDataParallel(delay_allreduce=True)``
(reference train_distributed.py:141-146), which launches ONE flat all-reduce
after backward finishes. On MI355X the xGMI fabric is point-to-point (7 links x
~153 GB/s per GPU), so communication time is best hidden under backward compute:
gradients are packed into fixed flat buckets in reverse-parameter (≈ backward
completion) order and each bucket's all-reduce is launched asynchronously the
moment its last gradient lands, via per-parameter post-accumulate hooks.

Key properties:
  * parameter ``.grad`` tensors are VIEWS into the flat bucket buffers — no
    pack/unpack copies, and the fused multi-tensor SGD step can consume the
    same flat buffers.
  * works over any torch.distributed backend: RCCL ("nccl") on MI355X,
    gloo on CPU for the multi-process unit tests.
  * bucket size defaults to 50 MiB — sized for per-link xGMI bandwidth
    (~153 GB/s => ~0.3 ms/bucket) rather than NVSwitch-era defaults.
"""
from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist


class Bucket:
    def __init__(self, params: List[torch.nn.Parameter], device, comm_dtype):
        self.params = params
        numel = sum(p.numel() for p in params)
        self.flat = torch.zeros(numel, dtype=comm_dtype, device=device)
        self.views = []
        offset = 0
        for p in params:
            v = self.flat[offset:offset + p.numel()].view_as(p)
            self.views.append(v)
            offset += p.numel()
        self.pending = 0
        self.work = None


class GradReducer:
    """Overlapped bucketed gradient all-reducer.

    Usage::
        reducer = GradReducer(model)          # after model is on its device
        for step:
            reducer.zero_grad()
            loss.backward()                    # buckets all-reduce as they fill
            reducer.finalize()                 # wait + average
            optimizer.step()
    """

    def __init__(self, module: torch.nn.Module, process_group=None,
                 bucket_cap_mb: float = 50.0, comm_dtype=None,
                 broadcast_parameters: bool = True):
        self.module = module
        self.group = process_group
        self.world_size = dist.get_world_size(process_group) if dist.is_initialized() else 1
        params = [p for p in module.parameters() if p.requires_grad]
        if not params:
            raise ValueError("model has no trainable parameters")
        device = params[0].device
        # comm_dtype is accepted for API stability but gradients must live in
        # each parameter's own dtype (the .grad views alias the flat buckets)
        del comm_dtype

        if self.world_size > 1 and broadcast_parameters:
            for p in params:
                dist.broadcast(p.data, src=0, group=self.group)
            for b in module.buffers():
                if b.dtype.is_floating_point or b.dtype in (torch.int64, torch.int32):
                    dist.broadcast(b.data, src=0, group=self.group)

        # reverse order ≈ order gradients become ready during backward.
        # Buckets are PER-DTYPE: a bf16 model still carries fp32 BN affine
        # parameters, and a parameter's .grad view must match its dtype.
        cap = int(bucket_cap_mb * 1024 * 1024)
        self.buckets: List[Bucket] = []
        cur: dict = {}
        cur_bytes: dict = {}
        for p in reversed(params):
            dt = p.dtype
            cur.setdefault(dt, []).append(p)
            cur_bytes[dt] = cur_bytes.get(dt, 0) + p.numel() * p.element_size()
            if cur_bytes[dt] >= cap:
                self.buckets.append(Bucket(cur[dt], device, dt))
                cur[dt], cur_bytes[dt] = [], 0
        for dt, ps in cur.items():
            if ps:
                self.buckets.append(Bucket(ps, device, dt))

        self._param_bucket = {}
        for b in self.buckets:
            for p, v in zip(b.params, b.views):
                p.grad = v  # autograd accumulates in-place into the bucket
                self._param_bucket[id(p)] = b
        self._hooks = [p.register_post_accumulate_grad_hook(self._on_grad)
                       for p in params]
        self.require_sync = True
        # IBP_DDP_TIMING=1: measure exposed (non-overlapped) all-reduce time
        # with CUDA events — evidence for the backward/communication overlap
        import os
        self.timing = (os.environ.get("IBP_DDP_TIMING") == "1"
                       and device.type == "cuda")
        self.last_timing = None
        self._reset_pending()

    # ------------------------------------------------------------------ hooks
    def _reset_pending(self):
        for b in self.buckets:
            b.pending = len(b.params)
            b.work = None
            b.launch_ev = None

    def _on_grad(self, param):
        if not self.require_sync or self.world_size <= 1:
            return
        b = self._param_bucket[id(param)]
        b.pending -= 1
        if b.pending == 0:
            if self.timing:
                b.launch_ev = torch.cuda.Event(enable_timing=True)
                b.launch_ev.record()
            b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                     group=self.group, async_op=True)

    # ------------------------------------------------------------------ public
    def zero_grad(self):
        for b in self.buckets:
            b.flat.zero_()
        self._reset_pending()

    def finalize(self):
        """Wait for in-flight all-reduces and average. Call after backward()."""
        if self.world_size <= 1 or not self.require_sync:
            return
        inv = 1.0 / self.world_size
        t_back = None
        if self.timing:
            t_back = torch.cuda.Event(enable_timing=True)
            t_back.record()  # ~end of backward compute
        for b in self.buckets:
            if b.work is None and b.pending > 0:
                # a parameter did not receive a gradient this step (e.g. an
                # unused head) — reduce the bucket anyway so ranks stay in step
                b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                         group=self.group, async_op=True)
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
            b.flat.mul_(inv)
        if self.timing:
            t_done = torch.cuda.Event(enable_timing=True)
            t_done.record()
            torch.cuda.synchronize()
            first = None
            for b in self.buckets:
                if b.launch_ev is not None:
                    first = b.launch_ev
                    break
            self.last_timing = {
                # collectives outstanding after backward finished = EXPOSED
                "exposed_ms": t_back.elapsed_time(t_done),
                # first bucket launch -> all reduced = total comm span
                # (span >> exposed means the overlap is doing its job)
                "comm_span_ms": (first.elapsed_time(t_done)
                                 if first is not None else 0.0),
                "buckets": len(self.buckets),
            }

    def flat_grads(self):
        """The flat bucket buffers (for the fused multi-tensor optimizer)."""
        return [b.flat for b in self.buckets]

    def detach(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []


def reduce_tensor(tensor: torch.Tensor, group=None) -> torch.Tensor:
    """Mean all-reduce for metric logging (reference train_distributed.py:428-438)."""
    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return tensor
    rt = tensor.clone()
    dist.all_reduce(rt, op=dist.ReduceOp.SUM, group=group)
    rt /= dist.get_world_size(group)
    return rt
