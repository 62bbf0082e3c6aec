"""Communication layer: RCCL (torch.distributed "nccl" backend on ROCm)
over xGMI for GPU runs, gloo for host-only CI, and a no-op single-process
fallback.

The reference has no in-repo comms (in-process method calls,
SURVEY.md 2b); this module realizes the collective call sites listed
there: boundary-pose exchange (packed all-gather — traffic is tiny, a
few 10s of KB, so a single latency-bound collective beats per-neighbor
p2p bookkeeping), status exchange, gradient-norm argmax for greedy
selection, termination consensus, and the anchor broadcast.
"""
from __future__ import annotations

import datetime
import os
from typing import List

import torch
import torch.distributed as dist

Tensor = torch.Tensor


class Comm:
    """Single-process no-op communicator (world size 1)."""

    rank = 0
    world_size = 1

    def all_gather_flat(self, local: Tensor, sizes: List[int]) -> List[Tensor]:
        return [local]

    def all_reduce_sum_(self, t: Tensor) -> Tensor:
        return t

    def barrier(self) -> None:
        pass


class TorchDistComm(Comm):
    """torch.distributed communicator (nccl=RCCL on GPU, gloo on CPU)."""

    def __init__(self, device: str = "cpu"):
        assert dist.is_initialized(), "torch.distributed not initialized"
        self.rank = dist.get_rank()
        self.world_size = dist.get_world_size()
        self.device = torch.device(device)
        self._backend = dist.get_backend()

    def all_gather_flat(self, local: Tensor, sizes: List[int]) -> List[Tensor]:
        """All-gather variable-size fp64 flats (sizes known per rank).
        Collectives need equal sizes -> pad to max (payloads are tiny).
        Send/recv buffers are persistent (keyed by max size) so the
        per-round path allocates nothing and RCCL sees stable pointers."""
        mx = max(sizes)
        cache = getattr(self, "_ag_cache", None)
        if cache is None or cache[0] != mx or cache[1].dtype != local.dtype:
            buf = torch.zeros(mx, dtype=local.dtype, device=self.device)
            out = torch.empty(self.world_size * mx, dtype=local.dtype,
                              device=self.device)
            self._ag_cache = cache = (mx, buf, out)
        _, buf, out = cache
        n = local.numel()
        buf[:n] = local.to(self.device)
        if n < mx:
            buf[n:] = 0.0
        dist.all_gather_into_tensor(out, buf)
        return [out[i * mx:i * mx + s] for i, s in enumerate(sizes)]

    def all_reduce_sum_(self, t: Tensor) -> Tensor:
        tt = t.to(self.device)
        dist.all_reduce(tt, op=dist.ReduceOp.SUM)
        if tt is not t:
            t.copy_(tt.to(t.device))
        return t

    def barrier(self) -> None:
        if self._backend == "nccl":
            dist.barrier(device_ids=[self.device.index or 0])
        else:
            dist.barrier()


def init_from_env(device: str = "cpu") -> Comm:
    """Initialize torch.distributed from torchrun env vars, or return the
    no-op communicator when not running under a launcher."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return Comm()
    if not dist.is_initialized():
        backend = "nccl" if device.startswith("cuda") else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
        dist.init_process_group(
            backend=backend,
            timeout=datetime.timedelta(seconds=300))
    return TorchDistComm(device)
