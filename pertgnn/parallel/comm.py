"""Comm — thin collective-communication interface.

The reference has NO distributed layer (single process, pert_gnn.py:36-37);
this layer is built to the MI355X north-star: one process per GPU,
``torch.distributed`` with backend "nccl" (= RCCL on ROCm) over xGMI inside a
node, "gloo" for CPU tests, and a no-op implementation for single-process
runs (SURVEY.md §5 "Distributed communication backend").
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


class Comm:
    """Process-group wrapper; a world_size-1 instance is a no-op."""

    def __init__(self, backend: str | None = None, timeout_s: int = 300):
        self.rank = int(os.environ.get("RANK", "0"))
        self.world_size = int(os.environ.get("WORLD_SIZE", "1"))
        self.local_rank = int(os.environ.get("LOCAL_RANK", str(self.rank)))
        self.distributed = self.world_size > 1
        if self.distributed and not dist.is_initialized():
            if backend is None:
                backend = "nccl" if torch.cuda.is_available() else "gloo"
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29500")
            dist.init_process_group(
                backend=backend,
                rank=self.rank,
                world_size=self.world_size,
                timeout=datetime.timedelta(seconds=timeout_s),
            )
        self.backend = dist.get_backend() if self.distributed else "none"
        if torch.cuda.is_available():
            torch.cuda.set_device(self.local_rank % torch.cuda.device_count())
            self.device = torch.device("cuda", self.local_rank % torch.cuda.device_count())
        else:
            self.device = torch.device("cpu")

    # -- collectives (no-ops at world_size 1) --------------------------------
    def all_reduce_(self, t: torch.Tensor, op: str = "sum", async_op: bool = False):
        if not self.distributed:
            return None
        red = {"sum": dist.ReduceOp.SUM, "max": dist.ReduceOp.MAX,
               "min": dist.ReduceOp.MIN}[op]
        return dist.all_reduce(t, op=red, async_op=async_op)

    def broadcast_(self, t: torch.Tensor, src: int = 0):
        if self.distributed:
            dist.broadcast(t, src=src)

    def barrier(self):
        if self.distributed:
            dist.barrier()

    def all_reduce_scalar(self, value: float, op: str = "sum") -> float:
        if not self.distributed:
            return value
        t = torch.tensor([value], dtype=torch.float64, device=self.device if self.backend == "nccl" else "cpu")
        self.all_reduce_(t, op=op)
        return float(t.item())

    def broadcast_module_(self, module: torch.nn.Module, src: int = 0):
        """Broadcast initial weights + buffers from rank src.

        Iterates parameters()/buffers() directly — NOT state_dict(), whose
        values can be detached re-mapped clones (the conv's fused w4/b4 are
        exported under the reference's per-projection keys) that an in-place
        broadcast would silently not write back."""
        if not self.distributed:
            return
        import itertools
        for p in itertools.chain(module.parameters(), module.buffers()):
            # skip lazily-uninitialized params (the model's dead edge_linear)
            if isinstance(p, torch.nn.parameter.UninitializedParameter) or \
               isinstance(p, torch.nn.parameter.UninitializedBuffer):
                continue
            if torch.is_tensor(p) and p.numel() > 0:
                self.broadcast_(p.data, src=src)

    def shard(self, seq):
        """Round-robin shard of a sequence for this rank (per-rank dataset
        sharding with identical global-batch semantics)."""
        return seq[self.rank::self.world_size]

    def finalize(self):
        if self.distributed and dist.is_initialized():
            dist.destroy_process_group()
