"""Bucketed data-parallel gradient all-reduce over RCCL/xGMI.

Works on the flat gradient buffers from FlatParams: buckets are contiguous
slices of the flat buffer taken in reverse layout order (the order grads
become ready during backward), reduced asynchronously as soon as every
param in the bucket has accumulated — overlapping communication with the
rest of backward.

xGMI tuning: each MI355X has 7 point-to-point links (~153 GB/s each); ring
all-reduce is per-link bound, so fewer/larger buckets amortize better than
the NVSwitch-tuned 25 MB default — we default to 64 MB
(TORCHX_AMD_BUCKET_MB overrides).  Backend "nccl" IS RCCL on ROCm; CPU CI
uses gloo (which lacks AVG, so we reduce SUM and divide once at finish).
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Set

import torch
import torch.distributed as dist

from .flat import FlatParams


# params whose grads are produced by DIRECT in-place accumulation (e.g.
# fast_linear's wgrad addmm_ into the flat slot) never trigger torch's
# post-accumulate hook; the producer calls grad_ready(param) instead so
# bucket overlap still works.  id(param) -> FlatDDP.
_PARAM_DDP: Dict[int, "FlatDDP"] = {}


def grad_ready(p: torch.Tensor) -> None:
    ddp = _PARAM_DDP.get(id(p))
    if ddp is not None:
        ddp._hook(p)


@dataclass
class _Bucket:
    group: str
    start: int
    end: int
    param_ids: Set[int] = field(default_factory=set)
    pending: int = 0
    launched: bool = False
    work: Optional[object] = None


class FlatDDP:
    def __init__(
        self,
        flat: FlatParams,
        process_group: Optional[dist.ProcessGroup] = None,
        bucket_bytes: Optional[int] = None,
        local_groups: Optional[set] = None,
    ):
        """``local_groups``: flat-param groups that are sharded (e.g. EP
        experts) — never all-reduced or broadcast, but still divided by
        world size in finish() so their grads match the dense params'
        (global objective = mean over ranks of local losses)."""
        self.flat = flat
        self.pg = process_group
        self.local_groups = local_groups or set()
        self.enabled = dist.is_available() and dist.is_initialized() and (
            dist.get_world_size(process_group) > 1
        )
        if bucket_bytes is None:
            bucket_bytes = int(os.environ.get("TORCHX_AMD_BUCKET_MB", "64")) << 20
        self.bucket_bytes = bucket_bytes
        # RCCL supports AVG in-collective: skips a whole-grad-buffer divide
        # sweep at finish (gloo lacks AVG -> reduce SUM + divide once)
        self.use_avg = (
            self.enabled and dist.get_backend(process_group) == "nccl"
        )
        self.sync_enabled = True
        self.buckets: List[_Bucket] = []
        self.param2bucket: Dict[int, _Bucket] = {}
        self._build_buckets()
        self._register_hooks()
        if self.enabled:
            self._broadcast_params()

    # -- setup --------------------------------------------------------------
    def _build_buckets(self) -> None:
        elem_size = 2  # bf16
        by_group: Dict[str, List] = {}
        for slot in self.flat.slots:
            if slot.group in self.local_groups:
                continue
            by_group.setdefault(slot.group, []).append(slot)
        for group, slots in by_group.items():
            cur: Optional[_Bucket] = None
            for slot in reversed(slots):  # backward-readiness order
                if cur is None:
                    cur = _Bucket(group=group, start=slot.offset,
                                  end=slot.offset + slot.numel)
                cur.start = min(cur.start, slot.offset)
                cur.end = max(cur.end, slot.offset + slot.numel)
                cur.param_ids.add(id(slot.param))
                self.param2bucket[id(slot.param)] = cur
                if (cur.end - cur.start) * elem_size >= self.bucket_bytes:
                    self.buckets.append(cur)
                    cur = None
            if cur is not None:
                self.buckets.append(cur)
        for b in self.buckets:
            b.pending = len(b.param_ids)

    def _register_hooks(self) -> None:
        for slot in self.flat.slots:
            slot.param.register_post_accumulate_grad_hook(self._hook)
            _PARAM_DDP[id(slot.param)] = self

    def _broadcast_params(self) -> None:
        for g, p16, _ in self.flat.groups():
            if g in self.local_groups:
                continue
            dist.broadcast(p16, src=0, group=self.pg)

    # -- runtime ------------------------------------------------------------
    def _hook(self, p: torch.Tensor) -> None:
        if not (self.enabled and self.sync_enabled):
            return
        b = self.param2bucket.get(id(p))
        if b is None:
            return
        b.pending -= 1
        if b.pending == 0:
            self._launch(b)

    def _launch(self, b: _Bucket) -> None:
        buf = self.flat.flat_grad[b.group][b.start:b.end]
        op = dist.ReduceOp.AVG if self.use_avg else dist.ReduceOp.SUM
        b.work = dist.all_reduce(buf, op=op, group=self.pg, async_op=True)
        b.launched = True

    def no_sync(self):
        """Context manager suppressing reduction (gradient accumulation)."""
        ddp = self

        class _Ctx:
            def __enter__(self):
                ddp.sync_enabled = False

            def __exit__(self, *a):
                ddp.sync_enabled = True

        return _Ctx()

    def finish(self) -> None:
        """Flush + wait all bucket reductions, then average.  Call between
        backward() and optimizer.step()."""
        if not (self.enabled and self.sync_enabled):
            self._reset()
            return
        for b in self.buckets:
            if not b.launched:
                self._launch(b)
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
        ws = dist.get_world_size(self.pg)
        for g, _, grad in self.flat.groups():
            # AVG path: reduced buckets are already averaged; only the
            # non-reduced (EP-local) groups still need the 1/ws factor
            if self.use_avg and g not in self.local_groups:
                continue
            grad.div_(ws)
        self._reset()

    def _reset(self) -> None:
        for b in self.buckets:
            b.pending = len(b.param_ids)
            b.launched = False
            b.work = None
