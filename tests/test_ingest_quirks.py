"""Unit tests for the reference's ingest quirks."""
import numpy as np
import pandas as pd
import torch

from pertgnn.data.graphs import build_pert_graph, build_span_graph, find_root_ms, sanitize_edges
from pertgnn.data.ingest import detect_entries


def _rows(rows):
    return pd.DataFrame(rows, columns=["traceid", "timestamp", "rpcid", "um", "rpctype", "dm", "interface", "rt"])


def test_entry_detection_tie_break_chain():
    """preprocess.py:111-131: http & earliest-ts & max|rt|; tie-break um='(?)';
    still ambiguous -> trace dropped; no candidate -> dropped."""
    df = _rows([
        # trace A: clean single candidate
        ("A", 100, "0", "(?)", "http", "svc1", "if1", -50),
        ("A", 101, "0.1", "svc1", "rpc", "svc2", "if2", 10),
        # trace B: two candidates at min ts/max rt, one with um='(?)' -> kept
        ("B", 100, "0", "(?)", "http", "svc1", "if1", 50),
        ("B", 100, "1", "svcX", "http", "svc3", "if3", 50),
        ("B", 102, "0.1", "svc1", "rpc", "svc2", "if2", 10),
        # trace C: two '(?)' candidates -> ambiguous, dropped
        ("C", 100, "0", "(?)", "http", "svc1", "if1", 50),
        ("C", 100, "1", "(?)", "http", "svc3", "if3", 50),
        # trace D: no http row -> dropped
        ("D", 100, "0", "(?)", "rpc", "svc1", "if1", 50),
    ])
    out = detect_entries(df)
    kept = set(out["traceid"])
    assert kept == {"A", "B"}
    assert set(out[out.traceid == "A"]["entryid"]) == {"svc1_if1"}
    assert set(out[out.traceid == "B"]["entryid"]) == {"svc1_if1"}


def test_sanitize_edges_heuristics():
    """misc.py:87-105 order: self-loops, dup rpcid keep-first, edges into
    root, dup (um,dm) keep-LAST, one edge per unordered pair keep-first."""
    df = pd.DataFrame({
        "traceid": ["T"] * 7,
        "timestamp": [100, 101, 102, 103, 104, 105, 106],
        "rpcid": ["0", "1", "1", "2", "3", "4", "5"],
        "um": [1, 2, 2, 3, 1, 2, 3],
        "rpctype": [0] * 7,
        "dm": [1, 3, 9, 1, 2, 1, 2],
        "interface": [10, 11, 12, 13, 14, 15, 16],
        "rt": [-99, 5, 6, 7, 8, 9, 10],
    })
    # root = um of max|rt| at min ts = row 0 -> um=1
    root = find_root_ms(df)
    assert root == 1
    out = sanitize_edges(df, root)
    pairs = list(zip(out["um"], out["dm"]))
    # row0 dropped (self-loop 1->1); row2 dropped (dup rpcid "1", keep first);
    # row3 (3->1) and row5 (2->1) dropped (dm == root);
    # remaining: (2,3) rpcid1, (1,2) rpcid3, (3,2) rpcid5;
    # unordered-pair dedup keeps FIRST of {1,2}? (1,2) at ts104 vs (3,2)... {3,2} vs {2,3}:
    # (2,3) and (3,2) same unordered pair -> keep first = (2,3)
    assert (2, 3) in pairs and (3, 2) not in pairs
    assert (1, 2) in pairs
    assert all(dm != root for dm in out["dm"])


def test_span_graph_relabels_consecutive():
    df = pd.DataFrame({
        "traceid": ["T"] * 3,
        "timestamp": [100, 101, 102],
        "rpcid": ["0", "1", "2"],
        "um": [7, 7, 20],
        "rpctype": [1, 0, 0],
        "dm": [20, 55, 55],
        "interface": [3, 4, 5],
        "rt": [-99, 5, 6],
    })
    g = build_span_graph(df)
    assert g["num_nodes"] == int(g["edge_index"].max()) + 1
    # ms ids preserved in sorted order
    assert g["ms_id"].flatten().tolist() == [7, 20, 55]
    assert g["edge_attr"].shape == (g["edge_index"].shape[1], 2)


def test_pert_stage_expansion_counts():
    """misc.py:240-257: caller with n calls -> 2n+1 stage nodes; pure callee -> 1."""
    df = pd.DataFrame({
        "traceid": ["T"] * 3,
        "timestamp": [100, 101, 103],
        "rpcid": ["0", "1", "2"],
        "um": [1, 1, 1],
        "rpctype": [0, 0, 0],
        "dm": [2, 3, 4],
        "interface": [9, 8, 7],
        "rt": [-99, 5, 6],
    })
    df["endTimestamp"] = df["timestamp"] + df["rt"].abs()
    g = build_pert_graph(df)
    # um=1 has 3 calls -> 7 stages; callees 2,3,4 -> 1 each => 10 nodes
    assert g["ms_id"].shape[0] == 10
    ms = g["ms_id"].flatten().tolist()
    assert ms[:7] == [1] * 7
    assert sorted(ms[7:]) == [2, 3, 4]
    # intra-ms chain: 6 edges attr [0,0,1,1]; plus 3 call + 3 return edges
    ea = g["edge_attr"]
    assert int(((ea[:, 2] == 1) & (ea[:, 3] == 1)).sum()) == 6
    assert int(((ea[:, 2] == 1) & (ea[:, 3] == 0)).sum()) == 3  # calls
    assert int(((ea[:, 2] == 0) & (ea[:, 3] == 0)).sum()) == 3  # returns


def test_y_is_max_abs_rt(synthetic_workspace):
    """preprocess.py:290-292: y = max |rt| over the trace."""
    root, (tr2data, *_rest) = synthetic_workspace
    import pandas as pd
    df = pd.read_csv(root / "processed" / "processed_df.csv")
    by_trace = df.groupby("traceid")["rt"].apply(lambda s: s.abs().max())
    for tid, d in list(tr2data.items())[:20]:
        assert float(d["y"]) == float(by_trace[tid])
