"""Hand-rolled bucketed gradient all-reduce over RCCL (gloo on CPU).

This is the MI355X-native replacement for the reference's Spark
parameter-server loop: instead of broadcast(weights) down
and collect/average up through the driver's Python heap every round, ranks
stay weight-synchronized and all-reduce GRADIENT BUCKETS over xGMI,
overlapped with the remaining backward.

Design (sized for the 8-GPU full-mesh xGMI topology, 7 p2p links x
~153 GB/s per GPU): parameters are grouped -- in reverse registration
order, approximating backward completion order -- into flat buckets of
`bucket_mb`. Each parameter's .grad is a VIEW into its bucket's flat
buffer, so backward accumulates in place. A post-accumulate-grad hook
counts readiness; when a bucket's last grad lands, an async all_reduce
launches immediately (on the communication stream under RCCL), overlapping
with the rest of backward. finalize() waits and averages.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


class _Bucket:
    def __init__(self, params: List[torch.nn.Parameter], device, dtype):
        self.params = params
        numel = sum(p.numel() for p in params)
        self.flat = torch.zeros(numel, dtype=dtype, device=device)
        self.views = []
        off = 0
        for p in params:
            self.views.append(self.flat[off : off + p.numel()].view_as(p))
            off += p.numel()
        self.pending = 0
        self.work: Optional[dist.Work] = None

    def numel(self) -> int:
        return self.flat.numel()


class GradBucketer:
    """Bucketed synchronous data-parallel gradient all-reduce.

    Usage per step:
        bucketer.zero_grad()
        loss.backward()          # hooks fire async all-reduces per bucket
        bucketer.finalize()      # wait + average
        optimizer.step()
    """

    def __init__(self, params, bucket_mb: int = 16,
                 process_group: Optional[dist.ProcessGroup] = None,
                 reduce_dtype: Optional[torch.dtype] = None,
                 bucket_order: str = "reverse"):
        """reduce_dtype: cast each bucket to this dtype for the wire
        (e.g. torch.float32 to avoid bf16 summation error across 8 ranks);
        None reduces in the native grad dtype.

        bucket_order: "reverse" groups parameters in reverse registration
        order (the transformer DDP heuristic).  "measured" additionally
        RECORDS the actual grad-arrival order during the first backward
        and rebuilds the buckets to match it -- a recurrent backward
        finishes all layers' dW near the END of its time loop (SURVEY §7
        hard part 3), so the registration heuristic can mis-group; the
        rebuild happens at the next zero_grad (grads are zeroed there
        anyway, so no gradient state is lost)."""
        self.pg = process_group
        self.reduce_dtype = reduce_dtype
        self.bucket_order = bucket_order
        self.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self.params = [p for p in params if p.requires_grad]
        if not self.params:
            raise ValueError("no parameters")
        self._device = self.params[0].device
        self._bucket_bytes = bucket_mb * (1 << 20)
        self._arrival: List[torch.nn.Parameter] = []
        self._arrival_ids: set = set()
        self._rebuilt = bucket_order != "measured"

        # Reverse order ~= backward completion order (last layers first).
        self._build_buckets(list(reversed(self.params)))
        self._hooks = [
            p.register_post_accumulate_grad_hook(self._on_grad_ready)
            for p in self.params
        ]
        self._reset_pending()

    def _build_buckets(self, ordered_params):
        self.buckets: List[_Bucket] = []
        cur: List[torch.nn.Parameter] = []
        cur_bytes = 0
        for p in ordered_params:
            esize = p.element_size()
            if cur and cur_bytes + p.numel() * esize > self._bucket_bytes:
                self.buckets.append(_Bucket(cur, self._device, cur[0].dtype))
                cur, cur_bytes = [], 0
            if cur and p.dtype != cur[0].dtype:
                self.buckets.append(_Bucket(cur, self._device, cur[0].dtype))
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += p.numel() * esize
        if cur:
            self.buckets.append(_Bucket(cur, self._device, cur[0].dtype))

        self._param_bucket = {}
        for b in self.buckets:
            for p, v in zip(b.params, b.views):
                self._param_bucket[p] = b
                p.grad = v  # accumulate directly into the bucket

    # ------------------------------------------------------------------
    def _reset_pending(self):
        for b in self.buckets:
            b.pending = len(b.params)
            b.work = None
        self._reduce_enabled = True

    def start_backward(self, will_reduce: bool = True):
        """Arm (or disarm) hook-triggered reduction for the next backward.

        With truncated BPTT, a step runs several backwards that accumulate
        into the same bucket buffers; only the LAST window's backward may
        launch the all-reduces (grads must be fully accumulated first).
        """
        for b in self.buckets:
            b.pending = len(b.params)
            b.work = None
        self._reduce_enabled = will_reduce

    def zero_grad(self):
        if not self._rebuilt and len(self._arrival) == len(self.params):
            # "measured" order: rebuild the buckets to the grad-arrival
            # order observed during the first backward (safe here: grads
            # are being zeroed anyway)
            self._build_buckets(self._arrival)
            self._rebuilt = True
        for b in self.buckets:
            b.flat.zero_()
        self._reset_pending()
        # re-attach grads as bucket views (optimizer.zero_grad may have
        # detached them by setting .grad = None)
        for b in self.buckets:
            for p, v in zip(b.params, b.views):
                p.grad = v

    def _on_grad_ready(self, p: torch.nn.Parameter):
        if not self._rebuilt and id(p) not in self._arrival_ids:
            self._arrival_ids.add(id(p))
            self._arrival.append(p)
        if self.world == 1 or not dist.is_initialized():
            return
        if not self._reduce_enabled:
            return
        b = self._param_bucket[p]
        b.pending -= 1
        if b.pending == 0:
            b.work = self._launch_reduce(b)

    def finalize(self):
        """Wait for all in-flight all-reduces and average."""
        if self.world == 1 or not dist.is_initialized():
            self._reset_pending()
            return
        inv = 1.0 / self.world
        for b in self.buckets:
            if b.work is None and b.pending > 0:
                # grads never produced this step (e.g. frozen path): reduce
                # anyway so ranks stay collective-aligned
                b.work = self._launch_reduce(b)
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
            if getattr(b, "wire", None) is not None:
                b.flat.copy_(b.wire)   # cast back from the fp32 wire buffer
                b.wire = None
            b.flat.mul_(inv)
        self._reset_pending()

    def _launch_reduce(self, b: "_Bucket"):
        if (self.reduce_dtype is not None
                and self.reduce_dtype != b.flat.dtype):
            b.wire = b.flat.to(self.reduce_dtype)
            return dist.all_reduce(b.wire, op=dist.ReduceOp.SUM,
                                   group=self.pg, async_op=True)
        b.wire = None
        return dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                               group=self.pg, async_op=True)

    def detach(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []

    def stats(self) -> dict:
        return {
            "num_buckets": len(self.buckets),
            "bucket_numel": [b.numel() for b in self.buckets],
            "total_mb": sum(b.numel() * b.flat.element_size()
                            for b in self.buckets) / (1 << 20),
        }


def broadcast_parameters(model: torch.nn.Module, src: int = 0,
                         process_group=None):
    """One-time weight broadcast from rank `src` at init (the reference's
    per-round broadcast collapses to this single call -- SURVEY.md §2.4)."""
    if not dist.is_initialized() or dist.get_world_size(process_group) == 1:
        return
    for p in model.parameters():
        dist.broadcast(p.data, src=src, group=process_group)
