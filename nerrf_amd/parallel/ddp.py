"""Bucketed, overlapped data-parallel gradient all-reduce (RCCL over xGMI).

Not torch DDP: we own the bucket schedule (SURVEY.md §7.3 — "do not lean on
DDP defaults").  On one MI355X node each GPU has 7 point-to-point xGMI links
(~153 GB/s each); a ring all-reduce is per-link bound, and this model's
gradients are small (~3 M params), so the schedule below is latency-oriented:

  * parameters are grouped in reverse creation order (the order grads become
    ready in backward) into few large buckets (default 8 MB — for a 3 M-param
    model that is 1-2 buckets, i.e. effectively one flat all-reduce, which is
    the right call when latency dominates bandwidth),
  * each bucket's flat buffer is persistent (no per-step allocation),
  * a bucket is reduced asynchronously the moment its last grad lands
    (post-accumulate-grad hooks), overlapping with the rest of backward,
  * `finalize()` waits and scatters the averaged grads back.

Process groups: init via `init_distributed()` — backend "nccl" IS RCCL on
ROCm; "gloo" for CPU tests (multi-process CPU CI runs world_size=2 gloo).
"""
from __future__ import annotations

import os
from typing import List, Optional

import torch
import torch.distributed as dist


def init_distributed(backend: Optional[str] = None) -> tuple[int, int, int]:
    """Initialise torch.distributed from torchrun env; returns (rank, world, local_rank)."""
    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        return 0, 1, 0
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if backend == "nccl":
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend=backend)
    return rank, world, local_rank


class GradAllReducer:
    """Bucketed async all-reduce of gradients, overlapped with backward.

    Contract: exactly ONE backward() per finalize().  A bucket is reduced
    the moment its last grad lands, so accumulating a second backward before
    finalize() would reduce the first pass's grads early; use a no-hook
    accumulation phase (or call finalize() per micro-batch) if micro-batching
    is ever added.
    """

    def __init__(
        self,
        model: torch.nn.Module,
        bucket_bytes: int = 8 << 20,
        process_group=None,
    ) -> None:
        self.group = process_group
        self.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self.enabled = self.world > 1
        params = [p for p in model.parameters() if p.requires_grad]
        params.reverse()  # grads become ready roughly in reverse creation order
        self.buckets: List[List[torch.nn.Parameter]] = []
        cur: List[torch.nn.Parameter] = []
        cur_bytes = 0
        for p in params:
            sz = p.numel() * p.element_size()
            if cur and cur_bytes + sz > bucket_bytes:
                self.buckets.append(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += sz
        if cur:
            self.buckets.append(cur)
        self._bucket_of = {}
        self._ready = [0] * len(self.buckets)
        self._filled = [False] * len(self.buckets)
        self._next_launch = 0
        self._flat: List[Optional[torch.Tensor]] = [None] * len(self.buckets)
        self._works: List = []
        if self.enabled:
            for bi, bucket in enumerate(self.buckets):
                for p in bucket:
                    self._bucket_of[p] = bi
                    p.register_post_accumulate_grad_hook(self._hook)

    def _hook(self, p: torch.nn.Parameter) -> None:
        bi = self._bucket_of[p]
        self._ready[bi] += 1
        if self._ready[bi] == len(self.buckets[bi]):
            self._filled[bi] = True
            self._drain()

    def _drain(self) -> None:
        # launch strictly in bucket-index order: with data-dependent heads
        # (a rank whose window has no sequences skips that head), arrival
        # order can differ across ranks, and out-of-order collectives on
        # one communicator hang or mismatch.  A filled bucket
        # waits until every lower-index bucket has launched; stragglers are
        # zero-filled and launched in order at finalize().
        while self._next_launch < len(self.buckets) and self._filled[self._next_launch]:
            self._launch(self._next_launch)
            self._next_launch += 1

    def _launch(self, bi: int) -> None:
        bucket = self.buckets[bi]
        grads = [p.grad for p in bucket]
        flat = torch._utils._flatten_dense_tensors(grads)
        work = dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=self.group, async_op=True)
        self._flat[bi] = flat
        self._works.append((work, bi))

    def finalize(self) -> None:
        """Wait on all pending reduces and write averaged grads back."""
        if not self.enabled:
            return
        # buckets whose hooks never all fired (e.g. a head unused this step):
        # zero-fill and launch in the same fixed index order as _drain
        for bi in range(self._next_launch, len(self.buckets)):
            if not self._filled[bi]:
                for p in self.buckets[bi]:
                    if p.grad is None:
                        p.grad = torch.zeros_like(p)
                self._filled[bi] = True
            self._drain()
        for work, bi in self._works:
            work.wait()
            flat = self._flat[bi]
            flat.div_(self.world)
            grads = [p.grad for p in self.buckets[bi]]
            for g, synced in zip(grads, torch._utils._unflatten_dense_tensors(flat, grads)):
                g.copy_(synced)
            self._flat[bi] = None
        self._works.clear()
        self._ready = [0] * len(self.buckets)
        self._filled = [False] * len(self.buckets)
        self._next_launch = 0

    def broadcast_params(self, model: torch.nn.Module, src: int = 0) -> None:
        if not self.enabled:
            return
        for p in model.parameters():
            dist.broadcast(p.data, src=src, group=self.group)
        for b in model.buffers():
            dist.broadcast(b.data, src=src, group=self.group)
