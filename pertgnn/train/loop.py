"""Training / evaluation loops (reference pert_gnn.py:213-294 semantics).

Differences from the reference that are performance-only (behavior-identical):
  * per-node rt_probs come precomputed from the collator instead of being
    rebuilt on CPU per batch (pert_gnn.py:220-230) — same values;
  * batches move to device with a single async H2D copy of the collated batch;
  * under DDP, gradient buckets all-reduce overlapped with backward and the
    summed metrics are all-reduced once per epoch.
"""
from __future__ import annotations

import torch

from ..ops import functional as F


class _nvtx:
    """roctx phase ranges (torch.cuda.nvtx maps to roctx on ROCm) — makes
    collate/H2D/fwd/bwd/allreduce/step phases visible in rocprofv3 traces."""

    def __init__(self, name):
        self.name = name
        self.on = torch.cuda.is_available()

    def __enter__(self):
        if self.on:
            torch.cuda.nvtx.range_push(self.name)

    def __exit__(self, *a):
        if self.on:
            torch.cuda.nvtx.range_pop()


def _forward(model, b):
    return model(
        b.x, b.cat_X, b.edge_index, b.edge_attr,
        b.pattern_num_nodes, b.rt_probs, b.entry_id, b.batch,
        csr=b.csr, num_graphs=b.num_graphs,
    )


def train_epoch(model, loader, optimizer, tau, device, engine=None, comm=None,
                non_blocking=True, stats_out=None, loss_scale=1.0):
    """Returns (avg_loss, avg_mape); with stats_out (dict) also records
    graphs/sec, edges/sec and nodes/sec for the epoch (whole job).
    ``loss_scale`` > 1 enables static loss scaling (fp16 mode): the scaled
    loss drives backward, the optimizer unscales (FusedAdam ``grad_scale``
    must match), and the REPORTED loss stays unscaled."""
    import time as _time

    model.train()
    n_graphs = 0
    n_edges = 0
    n_nodes = 0
    acc = None  # [loss*B sum, mape sum] accumulated ON DEVICE — the only
    # host sync is the one .tolist() after the epoch (reference semantics
    # float()'d per batch, pert_gnn.py:248-250, which serializes every step)
    t0 = _time.perf_counter()
    for batch in loader:
        with _nvtx("h2d"):
            b = batch.to(device, non_blocking=non_blocking) if device is not None else batch
        optimizer.zero_grad(set_to_none=False)
        if engine is not None:
            engine.reset()
        with _nvtx("forward"):
            global_pred, _local_pred = _forward(model, b)
            pred = global_pred.flatten()
            loss = F.quantile_loss(b.y, pred, tau)
        with _nvtx("backward"):
            # FusedAdam.scale_loss handles static AND dynamic loss scaling
            # (the dynamic scale is a device tensor — no host sync)
            if hasattr(optimizer, "scale_loss"):
                optimizer.scale_loss(loss).backward()
            elif loss_scale != 1.0:
                (loss * loss_scale).backward()
            else:
                loss.backward()
        with _nvtx("allreduce"):
            if engine is not None:
                engine.finalize()
        with _nvtx("optimizer"):
            optimizer.step()
        with torch.no_grad():
            if acc is None:
                acc = torch.zeros(2, dtype=torch.float64, device=pred.device)
            acc[0] += loss.detach().double() * b.num_graphs
            acc[1] += ((pred.detach() - b.y).abs() / b.y).sum().double()
            n_graphs += b.num_graphs
            n_edges += b.edge_index.shape[1]
            n_nodes += b.x.shape[0]
    if device is not None and torch.cuda.is_available():
        torch.cuda.synchronize()
    total_loss, mape_sum = (acc.tolist() if acc is not None else (0.0, 0.0))
    elapsed = _time.perf_counter() - t0
    if comm is not None and comm.distributed:
        total_loss = comm.all_reduce_scalar(total_loss)
        mape_sum = comm.all_reduce_scalar(mape_sum)
        n_graphs = int(comm.all_reduce_scalar(float(n_graphs)))
        n_edges = int(comm.all_reduce_scalar(float(n_edges)))
        n_nodes = int(comm.all_reduce_scalar(float(n_nodes)))
        elapsed = comm.all_reduce_scalar(elapsed, op="max")
    if stats_out is not None:
        stats_out.update({
            "epoch_s": elapsed,
            "graphs_per_s": n_graphs / max(elapsed, 1e-9),
            "edges_per_s": n_edges / max(elapsed, 1e-9),
            "nodes_per_s": n_nodes / max(elapsed, 1e-9),
        })
    n = max(n_graphs, 1)
    return total_loss / n, mape_sum / n


@torch.no_grad()
def evaluate(model, loader, tau, device, comm=None):
    model.eval()
    acc = None  # [mae, mape, qloss] sums on device; one sync per eval pass
    n_graphs = 0
    for batch in loader:
        b = batch.to(device) if device is not None else batch
        global_pred, _ = _forward(model, b)
        pred = global_pred.flatten()
        mae_s, mape_s, q_s = F.eval_metrics(b.y, pred, tau)
        if acc is None:
            acc = torch.zeros(3, dtype=torch.float64, device=pred.device)
        acc[0] += mae_s.double() if torch.is_tensor(mae_s) else mae_s
        acc[1] += mape_s.double() if torch.is_tensor(mape_s) else mape_s
        acc[2] += q_s.double() if torch.is_tensor(q_s) else q_s
        n_graphs += b.num_graphs
    mae, mape, qloss = (acc.tolist() if acc is not None else (0.0, 0.0, 0.0))
    if comm is not None and comm.distributed:
        mae = comm.all_reduce_scalar(mae)
        mape = comm.all_reduce_scalar(mape)
        qloss = comm.all_reduce_scalar(qloss)
        n_graphs = int(comm.all_reduce_scalar(float(n_graphs)))
    n = max(n_graphs, 1)
    return mae / n, mape / n, qloss / n
