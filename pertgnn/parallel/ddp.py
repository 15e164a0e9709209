"""Bucketed data-parallel gradient engine over RCCL/xGMI.

Design for the MI355X topology: the model is tiny (few-MB to
tens-of-MB gradients), xGMI is 7 point-to-point links per GPU, so all-reduce
latency — not bandwidth — dominates.  Therefore:

  * gradients are packed into a FEW large flat fp32 buckets (default 1,
    i.e. one fused all-reduce per step) instead of many small ones;
  * each bucket's all-reduce launches on a dedicated comm stream as soon as
    its last gradient is produced (post-accumulate hooks), overlapping the
    remaining backward;
  * ``finalize()`` syncs the comm stream, averages, and unpacks into
    ``param.grad`` before the optimizer step.

Works with any backend (gloo on CPU for tests — hooks then run the collective
inline since there are no streams).
"""
from __future__ import annotations

import torch

from .comm import Comm


class Bucket:
    def __init__(self, params, device, dtype=torch.float32):
        self.params = params
        numel = sum(p.numel() for p in params)
        self.flat = torch.zeros(numel, dtype=dtype, device=device)
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


class GradBucketAllReduce:
    """Attach to a model once; call ``reset()`` before backward and
    ``finalize()`` after backward, before optimizer.step()."""

    def __init__(self, model: torch.nn.Module, comm: Comm, bucket_cap_mb: float = 64.0):
        self.comm = comm
        self.params = [
            p for p in model.parameters()
            if p.requires_grad
            and not isinstance(p, torch.nn.parameter.UninitializedParameter)
        ]
        device = self.params[0].device if self.params else torch.device("cpu")
        self.use_stream = comm.device.type == "cuda"
        self.comm_stream = torch.cuda.Stream() if self.use_stream else None

        # bucket assignment in REVERSE parameter order (backward produces
        # gradients roughly last-to-first), capped at bucket_cap_mb
        cap = int(bucket_cap_mb * 1024 * 1024 / 4)
        self.buckets: list[Bucket] = []
        self.param2bucket: dict[int, tuple[Bucket, int]] = {}
        cur: list[torch.nn.Parameter] = []
        cur_numel = 0
        for p in reversed(self.params):
            cur.append(p)
            cur_numel += p.numel()
            if cur_numel >= cap:
                self._seal(cur, device)
                cur, cur_numel = [], 0
        if cur:
            self._seal(cur, device)

        for p in self.params:
            p.register_post_accumulate_grad_hook(self._hook)

    def _seal(self, params, device):
        b = Bucket(list(params), device)
        for i, p in enumerate(b.params):
            self.param2bucket[id(p)] = (b, i)
        self.buckets.append(b)

    def reset(self):
        for b in self.buckets:
            b.reset()

    def _hook(self, p: torch.nn.Parameter):
        if not self.comm.distributed:
            return
        b, i = self.param2bucket[id(p)]
        b.flat[b.offsets[i]: b.offsets[i] + p.numel()].copy_(p.grad.detach().reshape(-1))
        b.pending -= 1
        if b.pending == 0:
            if self.use_stream:
                self.comm_stream.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(self.comm_stream):
                    b.work = self.comm.all_reduce_(b.flat, async_op=True)
            else:
                b.work = self.comm.all_reduce_(b.flat, async_op=True)

    def finalize(self):
        if not self.comm.distributed:
            return
        # A bucket whose params did not ALL receive grads this step (e.g. the
        # dead local head outside the loss, quirk 3) never fired its hook:
        # launch its all-reduce now (zero slots for grad-less params — every
        # rank sees the same structure, so the collective schedule matches).
        for b in self.buckets:
            if b.work is None:
                if self.use_stream:
                    self.comm_stream.wait_stream(torch.cuda.current_stream())
                    with torch.cuda.stream(self.comm_stream):
                        b.work = self.comm.all_reduce_(b.flat, async_op=True)
                else:
                    b.work = self.comm.all_reduce_(b.flat, async_op=True)
        inv = 1.0 / self.comm.world_size
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
        if self.use_stream:
            torch.cuda.current_stream().wait_stream(self.comm_stream)
        for b in self.buckets:
            b.flat.mul_(inv)
            for i, p in enumerate(b.params):
                if p.grad is not None:
                    p.grad.detach().reshape(-1).copy_(b.flat[b.offsets[i]: b.offsets[i] + p.numel()])
            b.flat.zero_()
