"""Data parallelism: process-group bootstrap + bucketed gradient all-reduce.

The reference has no distributed support at all (SURVEY.md §2.3 — grep-clean of
nccl/gloo/DataParallel). Here: one process per GPU, torch.distributed with the
"nccl" backend (= RCCL over xGMI on ROCm) on GPU nodes, "gloo" on CPU for
tests. Instead of wrapping the model (which would prefix state_dict keys and
break checkpoint compatibility), gradients are all-reduced in flight via
post-accumulate-grad hooks with asynchronous ops so communication overlaps the
remaining backward — sized for the xGMI topology where a ring all-reduce is
bound by one ~153 GB/s link: MPGCN's gradient volume is small (O(100K-1M)
params; the model is activation-heavy), so latency, not bandwidth, dominates
and a single flat bucket per backward is the right shape.
"""

from __future__ import annotations

import contextlib
import os
import sys
import traceback
from dataclasses import dataclass
from datetime import timedelta

import torch
import torch.distributed as dist


@dataclass
class DistContext:
    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    backend: str = "none"

    @property
    def is_main(self) -> bool:
        return self.rank == 0

    @property
    def enabled(self) -> bool:
        return self.world_size > 1


def init_distributed(device: str = "cuda", backend: str | None = None,
                     timeout_s: int = 300) -> DistContext:
    """Initialize from torchrun env vars (RANK/WORLD_SIZE/LOCAL_RANK); no-op
    single-process context when they are absent. `timeout_s` bounds every
    collective: a dead or wedged rank turns into a clean RCCL timeout error
    instead of an indefinite hang (the failure-detection scope SURVEY.md §5
    prescribes for single-node 8-GPU)."""
    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        return DistContext()
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local = int(os.environ.get("LOCAL_RANK", rank))
    if world <= 1:
        return DistContext()
    if backend is None:
        backend = os.environ.get("MPGCN_DIST_BACKEND")  # test/bring-up override
    if backend is None:
        backend = "nccl" if (device.startswith("cuda") and torch.cuda.is_available()) else "gloo"
    if not dist.is_initialized():
        if backend == "nccl":
            # more ranks than GPUs (e.g. 2-rank RCCL bring-up on a 1-GPU box)
            # share devices round-robin
            torch.cuda.set_device(local % torch.cuda.device_count())
        dist.init_process_group(backend=backend, timeout=timedelta(seconds=timeout_s))
    return DistContext(rank=rank, world_size=world, local_rank=local, backend=backend)


@contextlib.contextmanager
def rank_watchdog(ctx: DistContext):
    """Clean-abort wrapper for distributed runs: on any exception, log the
    failing rank with its traceback, tear the process group down (so peers get
    a fast connection error instead of waiting out the collective timeout) and
    exit non-zero."""
    try:
        yield
    except Exception:
        sys.stderr.write(
            f"[mpgcn] rank {ctx.rank}/{ctx.world_size} failed:\n"
            + traceback.format_exc()
        )
        sys.stderr.flush()
        if ctx.enabled and dist.is_initialized():
            with contextlib.suppress(Exception):
                dist.destroy_process_group()
        if ctx.enabled:
            os._exit(1)
        raise


class GradAllReducer:
    """Bucketed asynchronous gradient all-reduce, hooked on the raw model.

    - broadcasts parameters from rank 0 at construction (identical init);
    - during backward, each parameter's grad joins the current bucket as it is
      accumulated; full buckets launch an async all-reduce immediately so
      communication overlaps the rest of backward;
    - `finalize()` flushes the tail bucket, waits for all in-flight reductions
      and averages (divide by world size).
    """

    def __init__(self, model: torch.nn.Module, ctx: DistContext,
                 bucket_bytes: int = 16 << 20):
        self.ctx = ctx
        self.model = model
        self.params = [p for p in model.parameters() if p.requires_grad]
        self.bucket_bytes = bucket_bytes
        self._pending: list[tuple[torch.distributed.Work, list[torch.Tensor], torch.Tensor]] = []
        self._bucket: list[torch.Tensor] = []
        self._bucket_sz = 0
        self._bucket_streams: set = set()  # streams that produced this bucket's grads
        self._hooks = []
        if not ctx.enabled:
            return
        with torch.no_grad():
            for p in self.params:
                dist.broadcast(p.data, src=0)
        for p in self.params:
            self._hooks.append(
                p.register_post_accumulate_grad_hook(self._on_grad_ready)
            )

    def _on_grad_ready(self, p: torch.Tensor):
        g = p.grad
        if g is None:
            return
        self._bucket.append(g)
        self._bucket_sz += g.numel() * g.element_size()
        if g.is_cuda:
            # autograd runs each AccumulateGrad on the stream its producing op
            # was recorded on (branch overlap records branches 1..M-1 on side
            # streams, models/mpgcn.py:155-180) — remember it so a mid-backward
            # flush can synchronize against every producer before reducing
            self._bucket_streams.add(torch.cuda.current_stream(g.device))
        if self._bucket_sz >= self.bucket_bytes:
            self._flush()

    def _flush(self):
        if not self._bucket:
            return
        grads = self._bucket
        streams = self._bucket_streams
        self._bucket = []
        self._bucket_sz = 0
        self._bucket_streams = set()
        if grads[0].is_cuda:
            # A bucket can mix gradients produced on different streams (branch
            # overlap). The flatten copies and the collective run on the
            # CURRENT stream, so order the current stream after every producing
            # stream first — without this, a bucket that fills mid-backward
            # launches all_reduce racing the side-stream grad writes.
            cur = torch.cuda.current_stream(grads[0].device)
            for s in streams:
                if s != cur:
                    cur.wait_stream(s)
        flat = torch._utils._flatten_dense_tensors(grads)
        if grads[0].is_cuda:
            # the flat buffer is consumed (all_reduce + unflatten-copy) on the
            # current stream; grads are later overwritten on it too — keep the
            # side-stream allocator from reusing their blocks early
            for g in grads:
                g.record_stream(torch.cuda.current_stream(grads[0].device))
        work = dist.all_reduce(flat, op=dist.ReduceOp.SUM, async_op=True)
        self._pending.append((work, grads, flat))

    def finalize(self):
        """Call between loss.backward() and optimizer.step()."""
        # torch guarantees grads are usable on the caller's stream once
        # backward() returns, but the model may have recorded branch work on
        # side streams (MPGCN branch overlap) — wait on them explicitly so the
        # ordering holds for any future call pattern.
        if torch.cuda.is_available():
            for s in getattr(self.model, "_streams", []):
                torch.cuda.current_stream().wait_stream(s)
        if not self.ctx.enabled:
            return
        self._flush()
        inv = 1.0 / self.ctx.world_size
        for work, grads, flat in self._pending:
            work.wait()
            flat.mul_(inv)
            for g, synced in zip(
                grads, torch._utils._unflatten_dense_tensors(flat, grads)
            ):
                g.copy_(synced)
        self._pending.clear()

    def remove(self):
        for h in self._hooks:
            h.remove()
        self._hooks.clear()
