"""Batch data parallelism over RCCL / xGMI — one process per GPU.

The reference has zero distributed code; this module is the
MI355X-native DP design:

  - torch.distributed with backend "nccl" (RCCL on ROCm) across the 8 GPUs of
    one node; "gloo" for CPU CI (the multi-process equivalence tests run on
    gloo, world_size 2, no GPU).
  - Gradients live in per-bucket FLAT buffers; parameter .grad tensors are
    views into them, so backward accumulates in place and the all-reduce is
    one contiguous collective per bucket — no per-tensor launches.
  - Buckets fill in reverse parameter order (the order backward produces
    grads) and each bucket's all_reduce launches asynchronously from a
    post-accumulate-grad hook, overlapping communication with the rest of
    backward. xGMI is 7 point-to-point links (~153 GB/s each); at ST-MGCN
    scale the gradient is ~1-25 MB, so the regime is latency-bound: default
    ONE bucket -> ONE collective per step. bucket_cap_mb
    tunes this for the large-city configs.
  - reduce() flushes un-launched buckets, waits all works, and rescales by
    1/world once on the flat storage.

Weights are broadcast from rank 0 at wrap time so all ranks start identical
(the reference-parity equivalent of its single-process init).
"""
from __future__ import annotations

import os
from typing import Dict, List, Optional

import torch
import torch.distributed as dist


def dist_env() -> Dict[str, int]:
    return {
        "rank": int(os.environ.get("RANK", "0")),
        "world_size": int(os.environ.get("WORLD_SIZE", "1")),
        "local_rank": int(os.environ.get("LOCAL_RANK", "0")),
    }


def init_distributed(backend: Optional[str] = None) -> Dict[str, int]:
    """Initialize torch.distributed from torchrun env vars. Returns the env
    dict; world_size 1 (no env) skips initialization entirely."""
    env = dist_env()
    if env["world_size"] <= 1:
        return env
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        if backend == "nccl":
            torch.cuda.set_device(env["local_rank"])
        dist.init_process_group(backend=backend)
    return env


def cleanup_distributed():
    if dist.is_initialized():
        dist.destroy_process_group()


class GradReducer:
    """Flat-bucket gradient all-reduce with backward overlap.

    Usage per step:
        reducer.zero_grad()
        loss.backward()          # hooks launch per-bucket async all_reduce
        reducer.reduce()         # flush + wait + 1/world rescale
        optimizer.step()
    """

    def __init__(self, model: torch.nn.Module, bucket_cap_mb: float = 25.0,
                 overlap: bool = True, process_group=None):
        self.params = [p for p in model.parameters() if p.requires_grad]
        self.group = process_group
        self.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self.overlap = overlap and self.world > 1
        cap = int(bucket_cap_mb * 1024 * 1024)

        # Reverse order: backward produces grads roughly last-to-first, so the
        # first-filled bucket completes earliest and its collective overlaps
        # the rest of backward.
        self.buckets: List[List[torch.nn.Parameter]] = []
        cur, cur_bytes = [], 0
        for p in reversed(self.params):
            nb = p.numel() * p.element_size()
            if cur and cur_bytes + nb > cap:
                self.buckets.append(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += nb
        if cur:
            self.buckets.append(cur)

        # flat buffer per bucket; p.grad = view
        self.flat: List[torch.Tensor] = []
        self._views: List[List[torch.Tensor]] = []
        for bucket in self.buckets:
            total = sum(p.numel() for p in bucket)
            buf = torch.zeros(total, dtype=bucket[0].dtype, device=bucket[0].device)
            views, ofs = [], 0
            for p in bucket:
                v = buf[ofs:ofs + p.numel()].view_as(p)
                p.grad = v
                views.append(v)
                ofs += p.numel()
            self.flat.append(buf)
            self._views.append(views)

        self._pending: List[Optional[dist.Work]] = [None] * len(self.buckets)
        self._arrived = [0] * len(self.buckets)
        self._param_bucket = {}
        for bi, bucket in enumerate(self.buckets):
            for p in bucket:
                self._param_bucket[p] = bi

        if self.overlap:
            for p in self.params:
                p.register_post_accumulate_grad_hook(self._on_grad_ready)

        if self.world > 1:  # identical initial weights on every rank
            with torch.no_grad():
                for p in model.parameters():
                    dist.broadcast(p.data, src=0, group=self.group)
            for b in model.buffers():
                dist.broadcast(b.data, src=0, group=self.group)

    # -------------------------------------------------------------- lifecycle
    def zero_grad(self):
        for buf in self.flat:
            buf.zero_()
        for bucket, views in zip(self.buckets, self._views):
            for p, v in zip(bucket, views):
                if p.grad is not v:   # re-pin if anything detached the view
                    p.grad = v
        self._pending = [None] * len(self.buckets)
        self._arrived = [0] * len(self.buckets)

    def _on_grad_ready(self, p: torch.nn.Parameter):
        bi = self._param_bucket[p]
        self._arrived[bi] += 1
        if self._arrived[bi] == len(self.buckets[bi]) and self._pending[bi] is None:
            self._pending[bi] = dist.all_reduce(self.flat[bi], group=self.group,
                                                async_op=True)

    def reduce(self):
        if self.world <= 1:
            return
        for bi in range(len(self.buckets)):
            if self._pending[bi] is None:
                self._pending[bi] = dist.all_reduce(self.flat[bi], group=self.group,
                                                    async_op=True)
        for w in self._pending:
            if w is not None:
                w.wait()
        inv = 1.0 / self.world
        for buf in self.flat:
            buf.mul_(inv)

    @property
    def grad_bytes(self) -> int:
        return sum(b.numel() * b.element_size() for b in self.flat)
