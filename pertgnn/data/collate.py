"""Batch collation (reference K17: PyG DataLoader concat, pert_gnn.py:196-210).

Produces a GraphBatch whose edges are DESTINATION-SORTED, with CSR/CSC index
arrays ready for the fused HIP edge-attention kernels — the collator, not the
kernels, pays the sorting cost (once per batch, on CPU, optionally via the
native C++ collator in csrc/collate.cpp when built).

Edge-order invariance: the model's math (segment softmax + segment sum) is
permutation-invariant over edges, so emitting edges dst-sorted is behaviorally
identical to the reference's concatenation order.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch

from .dataset import TraceSample


@dataclass
class GraphBatch:
    x: torch.Tensor                  # [N, F] float32
    cat_X: torch.Tensor              # [N, 1] int64
    edge_index: torch.Tensor         # [2, E] int64, sorted by dst
    edge_attr: torch.Tensor          # [E, A] int64 (CSR edge order)
    rt_probs: torch.Tensor           # [N, 1] float32
    pattern_num_nodes: torch.Tensor  # [N, 1] float32
    node_depth: torch.Tensor         # [N, 1] int64
    entry_id: torch.Tensor           # [B] int64
    batch: torch.Tensor              # [N] int64 graph id per node
    y: torch.Tensor                  # [B] float32
    # CSR (by dst) / CSC (by src) structure, int32
    row_ptr: torch.Tensor            # [N+1]
    csr_src: torch.Tensor            # [E]
    col_ptr: torch.Tensor            # [N+1]
    csc_dst: torch.Tensor            # [E]
    csc_eid: torch.Tensor            # [E] position into CSR edge order
    num_graphs: int
    batch_ptr: torch.Tensor | None = None  # [B+1] int32 (native collator)

    @property
    def csr(self):
        return (self.row_ptr, self.csr_src, self.col_ptr, self.csc_dst, self.csc_eid)

    def to(self, device, non_blocking: bool = False):
        kw = dict(device=device, non_blocking=non_blocking)
        return GraphBatch(
            x=self.x.to(**kw), cat_X=self.cat_X.to(**kw),
            edge_index=self.edge_index.to(**kw), edge_attr=self.edge_attr.to(**kw),
            rt_probs=self.rt_probs.to(**kw),
            pattern_num_nodes=self.pattern_num_nodes.to(**kw),
            node_depth=self.node_depth.to(**kw),
            entry_id=self.entry_id.to(**kw), batch=self.batch.to(**kw),
            y=self.y.to(**kw),
            row_ptr=self.row_ptr.to(**kw), csr_src=self.csr_src.to(**kw),
            col_ptr=self.col_ptr.to(**kw), csc_dst=self.csc_dst.to(**kw),
            csc_eid=self.csc_eid.to(**kw),
            num_graphs=self.num_graphs,
            batch_ptr=self.batch_ptr.to(**kw) if self.batch_ptr is not None else None,
        )

    def pin_memory(self):
        return GraphBatch(
            **{
                f: (getattr(self, f).pin_memory() if torch.is_tensor(getattr(self, f)) else getattr(self, f))
                for f in self.__dataclass_fields__
            }
        )


def collate_native(samples, pin: bool = False):
    """C++ single-pass collator (csrc/collate.cpp, reference K17): concat +
    edge offset + counting-sort CSR/CSC + batch vector in one native call,
    optionally into pinned host memory for async H2D."""
    from ..ops.backend import ext

    m = ext()
    if m is None or not hasattr(m, "collate_native"):
        return collate(samples)
    out = m.collate_native(
        [s.x for s in samples],
        [s.edge_index for s in samples],
        [s.edge_attr for s in samples],
        [s.cat_X for s in samples],
        [s.rt_probs for s in samples],
        [s.pattern_num_nodes for s in samples],
        [s.node_depth for s in samples],
        [s.entry_id for s in samples],
        [s.y for s in samples],
        pin,
    )
    (x, cat_X, edge_index, edge_attr, probs, pnn, nd, entry, batch,
     batch_ptr, y, row_ptr, csr_src, col_ptr, csc_dst, csc_eid) = out
    return GraphBatch(
        x=x, cat_X=cat_X, edge_index=edge_index, edge_attr=edge_attr,
        rt_probs=probs, pattern_num_nodes=pnn, node_depth=nd,
        entry_id=entry, batch=batch, y=y, row_ptr=row_ptr, csr_src=csr_src,
        col_ptr=col_ptr, csc_dst=csc_dst, csc_eid=csc_eid,
        num_graphs=len(samples), batch_ptr=batch_ptr,
    )


def build_csr(edge_index: torch.Tensor, num_nodes: int):
    """Sort edges by dst (stable) and build CSR + CSC index arrays.

    Returns (perm, row_ptr, csr_src, col_ptr, csc_dst, csc_eid): ``perm`` is
    the permutation mapping original edge order -> CSR order.
    """
    src, dst = edge_index[0], edge_index[1]
    perm = torch.argsort(dst, stable=True)
    src_s = src.index_select(0, perm)
    dst_s = dst.index_select(0, perm)
    row_counts = torch.bincount(dst_s, minlength=num_nodes)
    row_ptr = torch.zeros(num_nodes + 1, dtype=torch.int32)
    row_ptr[1:] = torch.cumsum(row_counts, 0).to(torch.int32)
    # CSC over the CSR-ordered edges
    perm2 = torch.argsort(src_s, stable=True)
    col_counts = torch.bincount(src_s, minlength=num_nodes)
    col_ptr = torch.zeros(num_nodes + 1, dtype=torch.int32)
    col_ptr[1:] = torch.cumsum(col_counts, 0).to(torch.int32)
    return (
        perm,
        row_ptr,
        src_s.to(torch.int32),
        col_ptr,
        dst_s.index_select(0, perm2).to(torch.int32),
        perm2.to(torch.int32),
    )


def collate(samples: list[TraceSample]) -> GraphBatch:
    """Concatenate samples into one disjoint-union batch graph."""
    node_offsets = []
    off = 0
    for s in samples:
        node_offsets.append(off)
        off += s.num_nodes
    num_nodes = off

    x = torch.cat([s.x for s in samples], dim=0)
    cat_X = torch.cat([s.cat_X for s in samples], dim=0)
    rt_probs = torch.cat([s.rt_probs for s in samples], dim=0)
    pattern_num_nodes = torch.cat([s.pattern_num_nodes for s in samples], dim=0)
    node_depth = torch.cat([s.node_depth for s in samples], dim=0)
    edge_index = torch.cat(
        [s.edge_index + o for s, o in zip(samples, node_offsets)], dim=1
    )
    edge_attr = torch.cat([s.edge_attr for s in samples], dim=0)
    entry_id = torch.cat([s.entry_id for s in samples])
    batch = torch.cat(
        [torch.full((s.num_nodes,), i, dtype=torch.long) for i, s in enumerate(samples)]
    )
    y = torch.stack([s.y.float() for s in samples])

    perm, row_ptr, csr_src, col_ptr, csc_dst, csc_eid = build_csr(edge_index, num_nodes)
    edge_index = edge_index.index_select(1, perm).contiguous()
    edge_attr = edge_attr.index_select(0, perm).contiguous()

    return GraphBatch(
        x=x, cat_X=cat_X, edge_index=edge_index, edge_attr=edge_attr,
        rt_probs=rt_probs, pattern_num_nodes=pattern_num_nodes,
        node_depth=node_depth, entry_id=entry_id, batch=batch, y=y,
        row_ptr=row_ptr, csr_src=csr_src, col_ptr=col_ptr,
        csc_dst=csc_dst, csc_eid=csc_eid, num_graphs=len(samples),
    )


class BatchLoader:
    """Minimal DataLoader over TraceSamples with reference split semantics
    (shuffle on train only, pert_gnn.py:201-209)."""

    def __init__(self, data_list, batch_size: int, shuffle: bool, seed: int = 0,
                 collate_fn=collate, drop_last: bool = False):
        self.data_list = data_list
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.generator = torch.Generator().manual_seed(seed)
        self.collate_fn = collate_fn
        self.drop_last = drop_last

    def __len__(self):
        n = len(self.data_list)
        if self.drop_last:
            return n // self.batch_size
        return (n + self.batch_size - 1) // self.batch_size

    @property
    def dataset(self):
        return self.data_list

    def __iter__(self):
        idx = (
            torch.randperm(len(self.data_list), generator=self.generator).tolist()
            if self.shuffle else range(len(self.data_list))
        )
        buf = []
        for i in idx:
            buf.append(self.data_list[i])
            if len(buf) == self.batch_size:
                yield self.collate_fn(buf)
                buf = []
        if buf and not self.drop_last:
            yield self.collate_fn(buf)
