"""Synthetic Alibaba-shaped call-graph trace generator.

Produces raw CSVs with the same schema the reference ingest consumes
(data/MSCallGraph/*.csv + data/MSResource/*.csv, reference
preprocess.py:203-242), shaped so every ingest filter passes by construction:

  * each trace has exactly one entry row: rpctype=='http', earliest
    timestamp, max |rt|, um=='(?)' (entry detection, preprocess.py:111-137)
  * resource rows exist for every 30 s bucket any trace references
    (quirk 14: find_most_recent_fts is an exact-match lookup, misc.py:373-376)
  * >=60% of each trace's microservices have resource rows (preprocess.py:155-177)
  * every entry has > 100 traces (preprocess.py:180-188) when
    traces_per_entry > 100

The generator is also used directly (in-memory) by bench.py to build
trace-scale batches without touching disk.
"""
from __future__ import annotations

import os
from dataclasses import dataclass, field

import numpy as np
import pandas as pd

from .schema import CALL_COLUMNS, RESOURCE_COLUMNS, TS_BUCKET_MS


@dataclass
class SyntheticConfig:
    n_entries: int = 4
    patterns_per_entry: int = 3
    traces_per_entry: int = 120
    # call-tree shape per runtime pattern
    min_calls: int = 4
    max_calls: int = 24
    max_fanout: int = 4
    n_microservices: int = 64
    n_interfaces: int = 40
    rpctypes: tuple = ("rpc", "mc", "db", "mq")
    resource_coverage: float = 0.9  # fraction of ms with resource rows
    base_latency_ms: int = 20
    seed: int = 0


@dataclass
class PatternSpec:
    """One runtime pattern: a call tree over microservices."""
    entry_ms: int
    entry_interface: int
    # list of (parent_slot, child_ms, interface, rpctype); slot 0 is the entry ms
    calls: list = field(default_factory=list)


def _make_pattern(rng: np.random.Generator, cfg: SyntheticConfig, entry_ms: int, entry_interface: int) -> PatternSpec:
    n_calls = int(rng.integers(cfg.min_calls, cfg.max_calls + 1))
    calls = []
    # nodes in the tree: ms ids; slot 0 = entry ms; children are distinct ms
    slots = [entry_ms]
    used = {entry_ms}
    for _ in range(n_calls):
        parent_slot = int(rng.integers(0, len(slots)))
        # cap fanout
        fanout = sum(1 for (p, *_rest) in calls if p == parent_slot)
        if fanout >= cfg.max_fanout:
            parent_slot = 0 if slots[0] is not None else parent_slot
        candidates = [m for m in range(cfg.n_microservices) if m not in used]
        if not candidates:
            break
        child = int(rng.choice(candidates))
        used.add(child)
        iface = int(rng.integers(0, cfg.n_interfaces))
        rpct = str(rng.choice(cfg.rpctypes))
        calls.append((parent_slot, child, iface, rpct))
        slots.append(child)
    return PatternSpec(entry_ms, entry_interface, calls)


def generate_patterns(cfg: SyntheticConfig) -> list[list[PatternSpec]]:
    rng = np.random.default_rng(cfg.seed)
    all_patterns = []
    for e in range(cfg.n_entries):
        entry_ms = int(rng.integers(0, cfg.n_microservices))
        entry_interface = int(rng.integers(0, cfg.n_interfaces))
        all_patterns.append(
            [_make_pattern(rng, cfg, entry_ms, entry_interface) for _ in range(cfg.patterns_per_entry)]
        )
    return all_patterns


def generate_traces(cfg: SyntheticConfig):
    """Returns (call_df, resource_df) raw dataframes with the Alibaba schema."""
    rng = np.random.default_rng(cfg.seed + 1)
    patterns = generate_patterns(cfg)

    call_rows = []
    ts_buckets = set()
    trace_counter = 0
    for e, entry_patterns in enumerate(patterns):
        probs = rng.dirichlet(np.ones(len(entry_patterns)) * 2.0)
        for _t in range(cfg.traces_per_entry):
            pat = entry_patterns[int(rng.choice(len(entry_patterns), p=probs))]
            traceid = f"trace_{trace_counter:08d}"
            trace_counter += 1
            t0 = int(rng.integers(0, 20)) * TS_BUCKET_MS + int(rng.integers(0, 1000))
            total_rt = int(cfg.base_latency_ms * (4 + len(pat.calls)) * float(rng.uniform(0.7, 1.4)))
            ts_buckets.add(t0 // TS_BUCKET_MS * TS_BUCKET_MS)
            # entry row: http, earliest ts, max |rt|, um == '(?)'
            call_rows.append(
                (traceid, t0, "0", "(?)", "http", f"MS_{pat.entry_ms}", f"IF_{pat.entry_interface}", total_rt)
            )
            # child calls: strictly later timestamps, strictly smaller rt
            slot_ms = [pat.entry_ms]
            for ci, (parent_slot, child, iface, rpct) in enumerate(pat.calls):
                # deterministic per-pattern call order: traces of one runtime
                # pattern produce identical um_dm_interface sequences (the
                # Alibaba data repeats patterns; only rt varies per trace)
                ts = t0 + 1 + 3 * ci
                rt = max(1, int(total_rt * float(rng.uniform(0.05, 0.5))))
                rt = min(rt, total_rt - 1)
                call_rows.append(
                    (traceid, ts, f"0.{ci + 1}", f"MS_{slot_ms[parent_slot]}", rpct,
                     f"MS_{child}", f"IF_{iface}", rt)
                )
                slot_ms.append(child)

    call_df = pd.DataFrame(call_rows, columns=CALL_COLUMNS)

    # resource rows: for every 30 s bucket any trace references, emit rows for
    # a cfg.resource_coverage fraction of microservices (several instances per
    # (ts, ms) so the max/min/mean/median aggregation is non-trivial)
    n_cov = max(1, int(round(cfg.resource_coverage * cfg.n_microservices)))
    covered_ms = rng.choice(cfg.n_microservices, size=n_cov, replace=False)
    res_rows = []
    for ts in sorted(ts_buckets):
        for ms in covered_ms:
            for _inst in range(3):
                res_rows.append(
                    (ts, f"MS_{ms}", float(rng.uniform(0.0, 1.0)), float(rng.uniform(0.0, 1.0)))
                )
    resource_df = pd.DataFrame(res_rows, columns=RESOURCE_COLUMNS)
    return call_df, resource_df


def write_dataset(root: str, cfg: SyntheticConfig | None = None):
    """Write data/MSCallGraph/*.csv + data/MSResource/*.csv under ``root``."""
    cfg = cfg or SyntheticConfig()
    call_df, resource_df = generate_traces(cfg)
    cg_dir = os.path.join(root, "data", "MSCallGraph")
    rs_dir = os.path.join(root, "data", "MSResource")
    os.makedirs(cg_dir, exist_ok=True)
    os.makedirs(rs_dir, exist_ok=True)
    call_df.to_csv(os.path.join(cg_dir, "MSCallGraph_0.csv"))
    resource_df.to_csv(os.path.join(rs_dir, "MSResource_0.csv"), index=False)
    return call_df, resource_df
