"""Data parallelism over RCCL/xGMI — bucketed all-reduce overlapped with
backward.

The reference has NO gradient-sync DP (its only multi-GPU use is
independent sweep agents, hyperparam_sweep/README.md:17); this is the
MI355X-native addition: one process per GPU, ``torch.distributed`` with the
nccl backend (= RCCL on ROCm), gradient buckets sized for the 7-link xGMI
topology (each MI355X has 7 p2p links x ~153 GB/s: a single ring is
per-link-bound, RCCL runs multi-ring when channels >= links, so buckets of
>= 64 MB keep every ring busy and amortize launch overhead — SURVEY.md §5
'Distributed communication backend').

Buckets are filled in REVERSE parameter order (last layer's grads are ready
first) and each bucket's all-reduce is issued asynchronously as soon as the
bucket is full, so communication of layer L overlaps with backward of
layer L-1. ``finalize()`` waits and writes averaged grads back.

Reduction dtype is fp32 regardless of parameter dtype: a bf16 ring-sum
across 8 ranks loses ~3 bits of mantissa on the 92 MB recurrent grads
(ring adds are serialized per hop), so grads are up-cast into an fp32
bucket buffer, summed in fp32 over the wire, averaged, and cast back to
each grad's own dtype on write-back. This also makes mixed-dtype buckets
(fp32 biases next to bf16 weights) lossless, so buckets need no dtype
partitioning. 2x wire bytes for bf16 params is the deliberate trade: the
all-reduce overlaps backward on xGMI and correctness of the sum wins.
``CI_DDP_BUCKET_MB`` overrides the bucket size for sweeps.

Deliberately not torch.nn.parallel.DistributedDataParallel: no graph
rewriting, no reducer C++ state — a small, inspectable bucketer suited to
this model family (few, large parameters: the 46 MB recurrent matrices and
the 96 MB tied embedding dominate).
"""
from __future__ import annotations

import os
from typing import List, Optional

import torch
import torch.distributed as dist
from torch import Tensor, nn

__all__ = ["init_distributed", "broadcast_parameters", "DistributedGrads"]


def init_distributed(backend: Optional[str] = None) -> tuple[int, int]:
    """Initialize from torchrun env; returns (rank, world_size).
    No-op (0,1) when not launched distributed."""
    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        return 0, 1
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        if backend == "nccl":
            # xGMI is 7 point-to-point links per GPU: a single-channel ring
            # is per-link-bound, so ask RCCL for at least 2 channels per
            # link (it clamps to what topology allows). setdefault keeps
            # any operator-provided tuning authoritative.
            os.environ.setdefault("NCCL_MIN_NCHANNELS", "14")
        dist.init_process_group(backend=backend)
        if torch.cuda.is_available():
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
    return dist.get_rank(), dist.get_world_size()


def broadcast_parameters(model: nn.Module, src: int = 0) -> None:
    if not (dist.is_available() and dist.is_initialized()):
        return
    with torch.no_grad():
        for p in model.state_dict().values():
            if isinstance(p, torch.Tensor) and p.numel():
                dist.broadcast(p, src)


class _Bucket:
    def __init__(self, params: List[Tensor], device):
        self.params = params
        self.numel = sum(p.numel() for p in params)
        # fp32 always: lossless sum for bf16/fp32 mixtures (see module doc)
        self.buffer = torch.empty(self.numel, device=device,
                                  dtype=torch.float32)
        self.offsets = []
        off = 0
        for p in params:
            self.offsets.append(off)
            off += p.numel()
        self.pending = 0
        self.work = None

    def reset(self):
        self.pending = len(self.params)
        self.work = None


class DistributedGrads:
    """Attach to a model once; call ``prepare()`` before each backward and
    ``finalize()`` after it (before optimizer.step())."""

    def __init__(self, model: nn.Module, bucket_mb: float = 64.0,
                 process_group=None):
        bucket_mb = float(os.environ.get("CI_DDP_BUCKET_MB", bucket_mb))
        self.enabled = dist.is_available() and dist.is_initialized() \
            and dist.get_world_size() > 1
        self.group = process_group
        self.world = dist.get_world_size() if self.enabled else 1
        params = [p for p in model.parameters() if p.requires_grad]
        # de-dup tied parameters (tied decoder/embedding share storage)
        seen, uniq = set(), []
        for p in params:
            if id(p) not in seen:
                seen.add(id(p))
                uniq.append(p)
        params = uniq
        self.buckets: List[_Bucket] = []
        self._param_bucket = {}
        self._armed = False  # hooks no-op unless prepare() armed this pass
        if not self.enabled:
            return
        cap = int(bucket_mb * 1024 * 1024)
        cur: List[Tensor] = []
        size = 0
        for p in reversed(params):  # reverse order: ready-first buckets
            cur.append(p)
            size += p.numel() * p.element_size()
            if size >= cap:
                self._mk_bucket(cur)
                cur, size = [], 0
        if cur:
            self._mk_bucket(cur)
        for b in self.buckets:
            for p, off in zip(b.params, b.offsets):
                self._param_bucket[id(p)] = (b, off)
                p.register_post_accumulate_grad_hook(self._hook)

    def _mk_bucket(self, params: List[Tensor]):
        self.buckets.append(_Bucket(params, params[0].device))

    def _hook(self, p: Tensor):
        if not self._armed:
            return  # gradient-accumulation micro-step: no communication
        b, off = self._param_bucket[id(p)]
        b.buffer[off: off + p.numel()].copy_(p.grad.reshape(-1))
        b.pending -= 1
        if b.pending == 0:
            b.work = dist.all_reduce(b.buffer, op=dist.ReduceOp.SUM,
                                     group=self.group, async_op=True)

    def prepare(self, sync: bool = True):
        """Arm the hooks for the coming backward. sync=False = this is a
        gradient-accumulation micro-step (DDP no_sync pattern): grads
        accumulate locally, no all-reduce fires."""
        self._armed = sync
        for b in self.buckets:
            b.reset()

    def finalize(self):
        if not self.enabled or not self._armed:
            return
        self._armed = False
        inv = 1.0 / self.world
        for b in self.buckets:
            if b.work is None:
                # some params got no grad this step (frozen/unused): fill
                # what exists, zero the rest, reduce synchronously
                for p, off in zip(b.params, b.offsets):
                    seg = b.buffer[off: off + p.numel()]
                    if p.grad is not None:
                        seg.copy_(p.grad.reshape(-1))
                    else:
                        seg.zero_()
                b.work = dist.all_reduce(b.buffer, op=dist.ReduceOp.SUM,
                                         group=self.group, async_op=True)
            if b.work is not None:
                b.work.wait()
            b.buffer.mul_(inv)
            for p, off in zip(b.params, b.offsets):
                if p.grad is not None:
                    p.grad.copy_(b.buffer[off: off + p.numel()].view_as(p.grad))
