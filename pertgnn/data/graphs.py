"""Span- and PERT-graph construction from a single trace's call rows.

Fresh MI355X-framework implementation of the behavior of the reference's
GraphConstruct (misc.py:72-370), numpy-first.  Behavioral contract
(SURVEY.md §2.1 rows "Graph sanitizer".."PERT-graph builder", §8 quirks):

  * root microservice = ``um`` of the row with max |rt| among rows at the
    minimum timestamp (misc.py:138-142), computed on the RAW trace rows.
  * edge sanitation order (misc.py:87-105): drop self-loops -> drop duplicate
    rpcid (keep first) -> drop edges into the root -> drop duplicate (um,dm)
    (keep LAST) -> keep one edge per unordered {um,dm} pair (keep first).
  * span graph (misc.py:190-219): nodes = sorted unique ms over the remaining
    (um,dm) rows, edges relabeled to consecutive ids; edge_attr = int64
    [interface, rpctype].
  * PERT graph (misc.py:221-370): each caller ms with n outgoing calls becomes
    a chain of 2n+1 stage nodes (intra-ms edges attr [0,0,1,1]); pure callees
    get 1 node; per caller its start/end events are time-sorted and wired
    start: stages[um][i] -> stages[dm][0]  attr [interface, rpctype, 1, 0]
    end:   stages[dm][-1] -> stages[um][i+1] attr [interface, rpctype, 0, 0].
  * node_depth: iterative min-depth DFS from the root with a monotone-decrease
    guard (tolerates residual cycles, misc.py:52-63); unreachable -> inf -> 0;
    normalized by the max depth and then CAST TO int64 (preprocess.py:338 via
    misc.py:155-173 + torch.long) — i.e. stored values are 0 except the
    deepest nodes (quirk: reproduced as-is).
"""
from __future__ import annotations

import numpy as np
import pandas as pd
import torch


def find_root_ms(trace_df: pd.DataFrame):
    """misc.py:138-142: um of (|rt|==max) & (ts==min) row, first occurrence."""
    rt_abs = trace_df["rt"].abs()
    mask = (rt_abs == rt_abs.max()) & (trace_df["timestamp"] == trace_df["timestamp"].min())
    sel = trace_df[mask]
    if len(sel) == 0:
        return None
    return sel["um"].iloc[0]


def sanitize_edges(trace_df: pd.DataFrame, root_ms) -> pd.DataFrame:
    """misc.py:87-105 semantics (cycle-break heuristics), same order."""
    df = trace_df[trace_df["um"] != trace_df["dm"]]
    df = df.drop_duplicates(subset="rpcid", keep="first")
    df = df[df["dm"] != root_ms]
    df = df.drop_duplicates(subset=["um", "dm"], keep="last")
    # one edge per unordered {um,dm} pair, keep first occurrence
    a = df["um"].to_numpy()
    b = df["dm"].to_numpy()
    lo = np.minimum(a, b)
    hi = np.maximum(a, b)
    key = pd.Series(list(zip(lo.tolist(), hi.tolist())), index=df.index)
    df = df[~key.duplicated(keep="first")]
    return df


def min_node_depth(num_nodes: int, edge_index: np.ndarray, root: int) -> np.ndarray:
    """Iterative version of misc.py:52-63 (min depth, monotone guard)."""
    if num_nodes == 0:
        return np.zeros(0)
    depth = np.full(num_nodes, np.inf)
    adj: list[list[int]] = [[] for _ in range(num_nodes)]
    for s, d in edge_index.T:
        adj[int(s)].append(int(d))
    stack = [(int(root), 0)]
    while stack:
        v, dep = stack.pop()
        if depth[v] > dep:
            depth[v] = dep
            for nb in adj[v]:
                stack.append((nb, dep + 1))
    return depth


def _normalized_depth_int64(depth: np.ndarray) -> torch.Tensor:
    if depth.size == 0:
        return torch.zeros(0, 1, dtype=torch.long)
    depth = depth.copy()
    depth[np.isinf(depth)] = 0
    norm = depth.max() if depth.max() > 0 else 1.0
    # reference stores torch.long of depth/norm (preprocess.py:338): truncation
    return torch.tensor((depth / norm)[:, None], dtype=torch.long)


def build_span_graph(trace_df: pd.DataFrame):
    """Returns dict(edge_index, ms_id, num_nodes, node_depth, edge_attr)
    matching the runtime2spangraph_map entry schema (preprocess.py:333-340)."""
    root_ms = find_root_ms(trace_df)
    df = sanitize_edges(trace_df, root_ms)
    um = df["um"].to_numpy(dtype=np.int64)
    dm = df["dm"].to_numpy(dtype=np.int64)
    pairs = np.stack([um, dm])  # [2,E]
    uniq, inv = np.unique(pairs, return_inverse=True)
    edge_index = torch.tensor(inv.reshape(2, -1), dtype=torch.long)
    num_nodes = int(edge_index.max().item()) + 1 if edge_index.numel() else 0
    ms2nid = {int(m): i for i, m in enumerate(uniq)}
    root_nid = ms2nid.get(int(root_ms), 0) if root_ms is not None else 0
    depth = min_node_depth(len(uniq), edge_index.numpy(), root_nid)
    edge_attr = torch.tensor(
        df[["interface", "rpctype"]].to_numpy(dtype=np.int64), dtype=torch.long
    )
    ms_id = torch.tensor(uniq[:, None], dtype=torch.long)
    return {
        "edge_index": edge_index.contiguous(),
        "ms_id": ms_id,
        "occurences": 1,
        "num_nodes": num_nodes,
        "node_depth": _normalized_depth_int64(depth),
        "edge_attr": edge_attr.contiguous(),
    }


def build_pert_graph(trace_df: pd.DataFrame):
    """Returns dict(edge_index, ms_id, num_nodes, node_depth, edge_attr[E,4])
    matching the runtime2pertgraph_map entry schema (preprocess.py:358-365)."""
    root_ms = find_root_ms(trace_df)
    df = sanitize_edges(trace_df, root_ms)
    if "endTimestamp" not in df.columns:
        df = df.assign(endTimestamp=df["timestamp"] + df["rt"].abs())

    stages: dict[int, np.ndarray] = {}
    sorted_ms_id: list[int] = []
    edges: list[tuple[int, int]] = []
    attrs: list[list[int]] = []
    num_nodes = 0
    # caller chains in value_counts order (misc.py:240: descending count,
    # ties by first appearance — pandas value_counts semantics)
    for um_ms, count in df["um"].value_counts().items():
        n_stages = 2 * int(count) + 1
        ids = np.arange(n_stages) + num_nodes
        stages[int(um_ms)] = ids
        for prev, cur in zip(ids[:-1], ids[1:]):
            edges.append((int(prev), int(cur)))
            attrs.append([0, 0, 1, 1])
        num_nodes += n_stages
        sorted_ms_id.extend([int(um_ms)] * n_stages)
    # pure-callee microservices get a single node (misc.py:251-257); the
    # reference iterates a Python set here (order unspecified) — we iterate
    # in first-appearance order of dm for determinism.
    um_set = set(df["um"].tolist())
    seen = set()
    for dm_ms in df["dm"].tolist():
        if dm_ms in um_set or dm_ms in seen:
            continue
        seen.add(dm_ms)
        stages[int(dm_ms)] = np.array([num_nodes])
        sorted_ms_id.append(int(dm_ms))
        num_nodes += 1

    # wire call/return edges per caller, events time-sorted (misc.py:272-302).
    # numpy throughout — the reference's per-row iterrows here is the
    # dominant cost of its 10+ hour preprocessing.
    um_arr = df["um"].to_numpy(dtype=np.int64)
    dm_arr = df["dm"].to_numpy(dtype=np.int64)
    ts_arr = df["timestamp"].to_numpy()
    ets_arr = df["endTimestamp"].to_numpy()
    ifc_arr = df["interface"].to_numpy(dtype=np.int64)
    rpc_arr = df["rpctype"].to_numpy(dtype=np.int64)
    order_um = np.argsort(um_arr, kind="stable")
    bounds = np.searchsorted(um_arr[order_um], np.unique(um_arr), side="left")
    uniq_um = np.unique(um_arr)
    for gi, um_ms in enumerate(uniq_um):
        lo = bounds[gi]
        hi = bounds[gi + 1] if gi + 1 < len(bounds) else len(order_um)
        rows = order_um[lo:hi]
        # events: (time, mode) with mode 0=start, 1=end; stable sort by time
        times = np.concatenate([ts_arr[rows], ets_arr[rows]])
        modes = np.concatenate([np.zeros(len(rows), dtype=np.int64),
                                np.ones(len(rows), dtype=np.int64)])
        dms = np.concatenate([dm_arr[rows], dm_arr[rows]])
        ifcs = np.concatenate([ifc_arr[rows], np.zeros(len(rows), dtype=np.int64)])
        rpcs = np.concatenate([rpc_arr[rows], np.zeros(len(rows), dtype=np.int64)])
        # reference interleaves (start,end) per row then stable-sorts by time:
        # replicate that insertion order before the sort
        interleave = np.empty(2 * len(rows), dtype=np.int64)
        interleave[0::2] = np.arange(len(rows))
        interleave[1::2] = np.arange(len(rows)) + len(rows)
        times = times[interleave]; modes = modes[interleave]
        dms = dms[interleave]; ifcs = ifcs[interleave]; rpcs = rpcs[interleave]
        ev_order = np.argsort(times, kind="stable")
        ids = stages[int(um_ms)]
        for i, e in enumerate(ev_order):
            dm_ms = int(dms[e])
            if modes[e] == 0:  # start: caller stage i -> callee first stage
                edges.append((int(ids[i]), int(stages[dm_ms][0])))
                attrs.append([int(ifcs[e]), int(rpcs[e]), 1, 0])
            else:  # end: callee last stage -> caller stage i+1
                edges.append((int(stages[dm_ms][-1]), int(ids[i + 1])))
                attrs.append([int(ifcs[e]), int(rpcs[e]), 0, 0])

    edge_index = torch.tensor(edges, dtype=torch.long).t().contiguous() if edges else torch.zeros(2, 0, dtype=torch.long)
    edge_attr = torch.tensor(attrs, dtype=torch.long).contiguous() if attrs else torch.zeros(0, 4, dtype=torch.long)
    nn_from_edges = int(edge_index.max().item()) + 1 if edge_index.numel() else 0
    root_nid = (int(stages[int(root_ms)][0])
                if root_ms is not None and int(root_ms) in stages else 0)
    depth = min_node_depth(num_nodes, edge_index.numpy(), root_nid)
    ms_id = torch.tensor(np.array(sorted_ms_id)[:, None], dtype=torch.long)
    return {
        "edge_index": edge_index,
        "ms_id": ms_id,
        "occurences": 1,
        # reference uses edge_index.max()+1 (preprocess.py:357)
        "num_nodes": nn_from_edges,
        "node_depth": _normalized_depth_int64(depth),
        "edge_attr": edge_attr,
    }
