"""Bucketed data-parallel gradient all-reduce over RCCL/xGMI.

The reference has NO distributed code (worker hardcodes
distributed=False, /root/reference/worker.py:481,490); the upstream 12-in-1
trainer used stock DDP over NCCL. This is the MI355X-native equivalent,
hand-rolled so the communication schedule is sized for the topology instead
of inherited:

- MI355X xGMI is point-to-point: 7 links x ~153 GB/s per GPU. A single-ring
  all-reduce is bound by ONE link, so RCCL needs buckets large enough to
  amortize per-collective launch cost but small enough that several are in
  flight across links while backward still computes. Default bucket:
  50 MB — the 270M-param model in bf16 grads (~540 MB) becomes ~11 buckets,
  each reduced asynchronously the moment its last gradient materializes
  (reverse parameter order), fully overlapped with backward.
- Works on the "nccl" backend (RCCL on ROCm) and "gloo" (CPU tests,
  world_size>1 — SURVEY.md §4 consequence (4)).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist


class _Bucket:
    def __init__(self, params: List[torch.nn.Parameter]):
        self.params = params
        self.pending = 0
        self.work: Optional[dist.Work] = None
        self.flat: Optional[torch.Tensor] = None
        self.grads: List[torch.Tensor] = []


class BucketedDataParallel:
    """Wraps a module already replicated across ranks (same init seed or an
    explicit broadcast) and overlaps gradient all-reduce with backward."""

    def __init__(
        self,
        module: torch.nn.Module,
        bucket_bytes: int = 50 * 1024 * 1024,
        process_group: Optional[dist.ProcessGroup] = None,
        broadcast_params: bool = True,
    ):
        self.module = module
        self.group = process_group
        self.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self.enabled = self.world > 1
        self._buckets: List[_Bucket] = []
        self._param_bucket: Dict[torch.nn.Parameter, _Bucket] = {}
        self._hooks = []
        # gradient accumulation: defer all reduction to finalize_backward()
        # (hook-time reduction would all-reduce PARTIAL micro-batch grads)
        self.defer_reduction = False
        if self.enabled:
            if broadcast_params:
                for p in module.parameters():
                    dist.broadcast(p.data, src=0, group=self.group)
            self._build_buckets(bucket_bytes)
            self._register_hooks()

    # ------------------------------------------------------------------
    def _build_buckets(self, bucket_bytes: int) -> None:
        # reverse order: grads materialize roughly output->input during
        # backward, so reversing parameter order lets early buckets fill
        # (and start reducing) first.
        params = [p for p in self.module.parameters() if p.requires_grad]
        params.reverse()
        cur: List[torch.nn.Parameter] = []
        size = 0
        for p in params:
            cur.append(p)
            size += p.numel() * p.element_size()
            if size >= bucket_bytes:
                self._buckets.append(_Bucket(cur))
                cur, size = [], 0
        if cur:
            self._buckets.append(_Bucket(cur))
        for b in self._buckets:
            for p in b.params:
                self._param_bucket[p] = b

    def _register_hooks(self) -> None:
        for p in self.module.parameters():
            if p.requires_grad:
                h = p.register_post_accumulate_grad_hook(self._on_grad)
                self._hooks.append(h)

    # ------------------------------------------------------------------
    def _on_grad(self, p: torch.nn.Parameter) -> None:
        if self.defer_reduction:
            return
        b = self._param_bucket[p]
        b.pending += 1
        if b.pending == len(b.params):
            self._reduce_bucket(b)

    def _reduce_bucket(self, b: _Bucket) -> None:
        # multi-task training: a step trains ONE task head, so params of the
        # other heads have no grad; all ranks run the same task per step
        # (rank-synchronized round-robin sampler), so skipping them is
        # collective-consistent.
        b.grads = [p.grad for p in b.params if p.grad is not None]
        if not b.grads:
            return
        b.flat = torch._utils._flatten_dense_tensors(b.grads)
        b.flat.div_(self.world)
        b.work = dist.all_reduce(b.flat, group=self.group, async_op=True)

    # ------------------------------------------------------------------
    def zero_grad(self, set_to_none: bool = True) -> None:
        self.module.zero_grad(set_to_none=set_to_none)
        for b in self._buckets:
            b.pending = 0
            b.work = None
            b.flat = None

    def finalize_backward(self) -> None:
        """Wait for in-flight reductions and scatter them back into .grad.
        Call between loss.backward() and optimizer.step()."""
        if not self.enabled:
            return
        for b in self._buckets:
            if b.work is None and (b.pending or self.defer_reduction):
                # deferred (grad accumulation), partially-filled bucket
                # (unused task heads), or late tail
                self._reduce_bucket(b)
        for b in self._buckets:
            if b.work is not None:
                b.work.wait()
                for g, r in zip(
                    b.grads, torch._utils._unflatten_dense_tensors(b.flat, b.grads)
                ):
                    g.copy_(r)
            b.pending = 0
            b.work = None
            b.flat = None
            b.grads = []

    # convenience passthroughs
    def __call__(self, *a, **kw):
        return self.module(*a, **kw)

    def parameters(self):
        return self.module.parameters()

    def state_dict(self):
        return self.module.state_dict()
