"""Async H2D batch prefetch (reference K16 — data.to(device) at
pert_gnn.py:231 re-designed as pinned staging + a dedicated copy stream).

The collator writes into pinned host memory (collate_native(pin=True)); this
loader starts the H2D copy of batch i+1 on a side stream while batch i
computes, so the device never waits on PCIe/IF for resident-sized batches.
"""
from __future__ import annotations

import torch


class PrefetchLoader:
    def __init__(self, loader, device):
        self.loader = loader
        self.device = device
        self.use_stream = (
            isinstance(device, torch.device) and device.type == "cuda"
        ) or (isinstance(device, str) and "cuda" in device)
        self.stream = torch.cuda.Stream() if self.use_stream else None

    def __len__(self):
        return len(self.loader)

    @property
    def dataset(self):
        return self.loader.dataset

    def _start_copy(self, batch):
        with torch.cuda.stream(self.stream):
            return batch.to(self.device, non_blocking=True)

    def __iter__(self):
        if not self.use_stream:
            yield from self.loader
            return
        it = iter(self.loader)
        host = next(it, None)
        pending = self._start_copy(host) if host is not None else None
        while pending is not None:
            nxt_host = next(it, None)
            torch.cuda.current_stream().wait_stream(self.stream)
            cur = pending
            # protect the prefetched tensors from premature reuse by the
            # side-stream allocator
            for f in cur.__dataclass_fields__:
                t = getattr(cur, f)
                if torch.is_tensor(t) and t.is_cuda:
                    t.record_stream(torch.cuda.current_stream())
            pending = self._start_copy(nxt_host) if nxt_host is not None else None
            yield cur


class ThreadedLoader:
    """Runs the underlying loader (collation) in a worker thread — the native
    collator releases the GIL, so batch i+1 is collated while batch i trains.
    Bounded queue (depth 2) keeps at most two host batches in flight."""

    def __init__(self, loader, depth: int = 2):
        self.loader = loader
        self.depth = depth

    def __len__(self):
        return len(self.loader)

    @property
    def dataset(self):
        return self.loader.dataset

    def __iter__(self):
        import queue
        import threading

        q: "queue.Queue" = queue.Queue(maxsize=self.depth)
        SENTINEL = object()
        err = []

        def worker():
            try:
                for b in self.loader:
                    q.put(b)
            except BaseException as e:  # propagate into the consumer
                err.append(e)
            finally:
                q.put(SENTINEL)

        t = threading.Thread(target=worker, daemon=True)
        t.start()
        while True:
            item = q.get()
            if item is SENTINEL:
                break
            yield item
        t.join()
        if err:
            raise err[0]
