"""Data-parallel gradient synchronization over RCCL / xGMI.

New first-class component (the reference is single-process — SURVEY.md 2.3):
one process per GPU, `torch.distributed` with the nccl backend (= RCCL on
ROCm). Two modes, chosen for the xGMI point-to-point topology (7 links x
~153 GB/s per GPU; ring collectives are per-link bound):

- "bucketed" (default): gradients are packed into flat buckets in reverse
  parameter order and all-reduced asynchronously as soon as each bucket's
  last gradient is produced — the leading buckets' communication overlaps
  the remaining backward compute. The default bucket size (1 MiB) is sized
  so the flagship model (~1.9 MB of fp32 gradients at L2 H128) splits into
  2 buckets, i.e. overlap actually exists rather than being nominal.
- "fused": ONE flat all-reduce over every gradient at finalize(). For tiny
  payloads the per-collective launch+rendezvous latency dominates transfer
  time, so a single collective with zero overlap can beat three overlapped
  ones; the bench can select whichever wins on the measured hardware.

Works with any backend (gloo for CPU tests, nccl/RCCL on MI355X).
"""
from typing import List, Optional

import torch
import torch.distributed as dist


class GradAllReduce:
    """Bucketed (or single-shot fused) gradient all-reduce.

    Usage per step:
        loss.backward()          # bucketed: hooks fire, buckets launch async
        engine.finalize()        # wait/launch + write averaged grads back
        clip / optimizer.step()
    """

    def __init__(self, model: torch.nn.Module, bucket_bytes: int = 1 << 20,
                 process_group=None, mode: str = "bucketed"):
        assert mode in ("bucketed", "fused"), mode
        self.pg = process_group
        self.mode = mode
        self.world = dist.get_world_size(process_group)
        self.params: List[torch.nn.Parameter] = [
            p for p in model.parameters() if p.requires_grad]

        # Broadcast initial parameters so every rank starts identical.
        with torch.no_grad():
            for p in self.params:
                dist.broadcast(p.data, src=0, group=self.pg)

        # Build buckets in reverse order (grads are produced roughly in
        # reverse registration order during backward). In fused mode there
        # is a single bucket and no hooks — nothing launches until
        # finalize(), where one collective covers the whole payload.
        if mode == "fused":
            self.buckets = [list(reversed(self.params))]
        else:
            self.buckets = self.build_buckets(self.params, bucket_bytes)

        self._bucket_of = {}
        self._flat: List[Optional[torch.Tensor]] = [None] * len(self.buckets)
        self._pending: List[int] = [0] * len(self.buckets)
        self._works: List[Optional[dist.Work]] = [None] * len(self.buckets)
        if mode == "bucketed":
            for bi, bucket in enumerate(self.buckets):
                for p in bucket:
                    self._bucket_of[p] = bi
                    p.register_post_accumulate_grad_hook(self._hook)
        self._reset()

    @staticmethod
    def build_buckets(params, bucket_bytes: int):
        """Reverse-order greedy bucketing (pure; unit-testable without a
        process group)."""
        buckets, cur, cur_bytes = [], [], 0
        for p in reversed(list(params)):
            cur.append(p)
            cur_bytes += p.numel() * p.element_size()
            if cur_bytes >= bucket_bytes:
                buckets.append(cur)
                cur, cur_bytes = [], 0
        if cur:
            buckets.append(cur)
        return buckets

    def _reset(self):
        for bi in range(len(self.buckets)):
            self._pending[bi] = len(self.buckets[bi])
            self._works[bi] = None
            self._flat[bi] = None

    def _launch(self, bi: int):
        bucket = self.buckets[bi]
        grads = [q.grad for q in bucket if q.grad is not None]
        if not grads:
            return
        flat = torch._utils._flatten_dense_tensors(grads)
        self._flat[bi] = flat
        self._works[bi] = dist.all_reduce(flat, op=dist.ReduceOp.SUM,
                                          group=self.pg, async_op=True)

    def _hook(self, p: torch.nn.Parameter):
        bi = self._bucket_of[p]
        self._pending[bi] -= 1
        if self._pending[bi] == 0:
            self._launch(bi)

    def finalize(self):
        """Wait for (or in fused mode, launch) the all-reduce(s) and write
        averaged gradients back into param.grad."""
        inv = 1.0 / self.world
        for bi, bucket in enumerate(self.buckets):
            if self._works[bi] is None:
                # fused mode launches here by design; in bucketed mode this
                # catches a bucket whose hook chain never completed (some
                # param unused this step) but that still holds gradients
                self._launch(bi)
            work = self._works[bi]
            if work is None:
                continue  # bucket had no grads this step
            work.wait()
            flat = self._flat[bi]
            flat.mul_(inv)
            with_grads = [q for q in bucket if q.grad is not None]
            for p, g in zip(with_grads, torch._utils._unflatten_dense_tensors(
                    flat, [q.grad for q in with_grads])):
                p.grad.copy_(g)
        self._reset()


def init_distributed(backend: Optional[str] = None) -> int:
    """Initialize torch.distributed from torchrun env vars; returns rank.
    Backend defaults to nccl (RCCL) when a GPU is visible, else gloo."""
    import os
    if dist.is_initialized():
        return dist.get_rank()
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29541")
    dist.init_process_group(backend=backend)
    return dist.get_rank()
