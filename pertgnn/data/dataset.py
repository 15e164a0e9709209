"""Dataset assembly (reference L2, pert_gnn.py:40-188): one sample per trace =
the disjoint union of ALL runtime patterns of the trace's entry.

Reference behavior reproduced exactly:
  * quirk 5: duplicate-ms PERT feature assignment collapses — only the LAST
    stage node of each microservice receives resource features; all other
    copies keep zeros + missing-indicator 1 (pert_gnn.py:56 dict-comprehension
    keeps the last nid).
  * per-node pattern probability (the ``rt_probs`` the training loop rebuilds
    per batch at pert_gnn.py:220-230) is precomputed per sample here — the
    values are identical, computed once instead of per epoch.
  * the per-pattern ``pattern_probs`` field is kept (dead in the reference
    loss path, quirk 3) for checkpoint/data compatibility.
"""
from __future__ import annotations

from dataclasses import dataclass
from functools import lru_cache

import numpy as np
import pandas as pd
import torch



@dataclass
class TraceSample:
    """Same tensor fields as the reference PyG Data (pert_gnn.py:163-173)."""
    x: torch.Tensor                  # [N, 9] float32
    edge_index: torch.Tensor         # [2, E] int64
    edge_attr: torch.Tensor          # [E, 2] (span) / [E, 4] (pert) int64
    cat_X: torch.Tensor              # [N, 1] int64 (ms id per node)
    node_depth: torch.Tensor         # [N, 1] int64
    pattern_num_nodes: torch.Tensor  # [N, 1] float32
    pattern_probs: torch.Tensor      # [P, 1] float32 (per pattern; dead, kept for compat)
    rt_probs: torch.Tensor           # [N, 1] float32 (per node; what the model consumes)
    entry_id: torch.Tensor           # [1] int64
    y: torch.Tensor                  # 0-d

    @property
    def num_nodes(self):
        return self.x.shape[0]

    @property
    def num_edges(self):
        return self.edge_index.shape[1]


class ResourceLookup:
    """Exact-match (timestamp, msname) -> 8 features (misc.py:373-376 is an
    exact .loc — a missing bucket raises, reproduced here as KeyError)."""

    def __init__(self, resource_df: pd.DataFrame):
        feats = resource_df.drop(columns=["timestamp", "msname"]).to_numpy(dtype=np.float32)
        ts = resource_df["timestamp"].to_numpy()
        ms = resource_df["msname"].to_numpy()
        self._map = {(int(t), int(m)): feats[i] for i, (t, m) in enumerate(zip(ts, ms))}
        self.ms_with_resources = frozenset(int(m) for m in np.unique(ms))
        self.num_features = feats.shape[1]

    def get(self, timestamp: int, ms: int):
        return self._map[(timestamp, ms)]

    def has(self, timestamp: int, ms: int) -> bool:
        return (timestamp, ms) in self._map


class EntryUnionCache:
    """Per-entry concatenation of all runtime patterns — the equivalent of the
    reference's six @lru_cache helpers (pert_gnn.py:77-131)."""

    def __init__(self, entry2runtimes: dict, runtime2graph: dict):
        self.entry2runtimes = entry2runtimes
        self.runtime2graph = runtime2graph

    @lru_cache(maxsize=None)
    def static_union(self, entry_id: int):
        rt_ids = list(self.entry2runtimes[entry_id].keys())
        rt_probs = list(self.entry2runtimes[entry_id].values())
        graphs = [self.runtime2graph[r] for r in rt_ids]
        nn = [g["num_nodes"] for g in graphs]
        offsets = np.concatenate([[0], np.cumsum(nn)[:-1]])
        edge_index = torch.cat(
            [g["edge_index"] + int(off) for g, off in zip(graphs, offsets)], dim=1
        )
        edge_attr = torch.cat([g["edge_attr"] for g in graphs], dim=0)
        cat_X = torch.cat([g["ms_id"] for g in graphs], dim=0)
        node_depth = torch.cat([g["node_depth"] for g in graphs], dim=0)
        pattern_num_nodes = torch.tensor(
            [[n] for n in nn for _ in range(n)], dtype=torch.float
        )
        rt_probs_node = torch.tensor(
            [[p] for p, n in zip(rt_probs, nn) for _ in range(n)], dtype=torch.float
        )
        pattern_probs = torch.tensor(np.array(rt_probs)[:, None], dtype=torch.float)
        # per-pattern flat ms list for feature lookup (pert_gnn.py:141-155)
        per_node_ms = [
            [int(m) for m in g["ms_id"].flatten().tolist()] for g in graphs
        ]
        return (edge_index, edge_attr, cat_X, node_depth, pattern_num_nodes,
                rt_probs_node, pattern_probs, per_node_ms)


def build_x(timestamp: int, per_node_ms: list[list[int]], res: ResourceLookup) -> torch.Tensor:
    """Reference get_x per pattern then concat (pert_gnn.py:40-67,141-155),
    including quirk 5 (keep-last nid for duplicated ms)."""
    blocks = []
    nf = res.num_features
    for ms_list in per_node_ms:
        n = len(ms_list)
        x = np.zeros((n, nf + 1), dtype=np.float32)
        x[:, nf] = 1.0  # missing indicator
        ms2nid = {ms: nid for nid, ms in enumerate(ms_list)}  # keeps LAST
        for ms in ms_list:
            if res.has(timestamp, ms):
                nid = ms2nid[ms]
                x[nid, :nf] = res.get(timestamp, ms)
                x[nid, nf] = 0.0
        blocks.append(x)
    return torch.tensor(np.concatenate(blocks, axis=0))


def build_data_list(tr2data: dict, entry2runtimes: dict, runtime2graph: dict,
                    resource_df: pd.DataFrame, limit: int | None = 100000,
                    verbose: bool = False) -> list[TraceSample]:
    """Reference get_data_list (pert_gnn.py:176-188) with a 100k-trace cap
    (pert_gnn.py:298-299)."""
    res = ResourceLookup(resource_df)
    cache = EntryUnionCache(entry2runtimes, runtime2graph)
    samples = []
    items = list(tr2data.items())
    if limit is not None:
        items = items[:limit]
    for traceid, data in items:
        entry_id = int(data["entry_id"])
        timestamp = int(data["timestamp"])
        (edge_index, edge_attr, cat_X, node_depth, pattern_num_nodes,
         rt_probs_node, pattern_probs, per_node_ms) = cache.static_union(entry_id)
        x = build_x(timestamp, per_node_ms, res)
        samples.append(
            TraceSample(
                x=x,
                edge_index=edge_index,
                edge_attr=edge_attr,
                cat_X=cat_X,
                node_depth=node_depth,
                pattern_num_nodes=pattern_num_nodes,
                pattern_probs=pattern_probs,
                rt_probs=rt_probs_node,
                entry_id=torch.tensor([entry_id], dtype=torch.long),
                y=data["y"],
            )
        )
    return samples


def split_60_20_20(data_list: list):
    """Sequential 60/20/20 split, no shuffle before splitting (quirk 7,
    pert_gnn.py:196-210)."""
    n = len(data_list)
    return (
        data_list[: int(n * 0.6)],
        data_list[int(n * 0.6): int(n * 0.8)],
        data_list[int(n * 0.8):],
    )
