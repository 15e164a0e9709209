"""End-to-end pipeline: synthetic traces -> ingest -> dataset -> collate -> model."""
import os
from pathlib import Path

import joblib
import pytest
import torch

from pertgnn.data.collate import BatchLoader, collate
from pertgnn.data.dataset import build_data_list, split_60_20_20
from pertgnn.models import SAGEDeterministic
from pertgnn.ops import functional as F


def test_artifact_schema(synthetic_workspace):
    root, (tr2data, entry2runtimes, runtime2span, runtime2pert, resource_df) = synthetic_workspace
    pdir = root / "processed"
    for f in ("tr2data.pt", "runtime2spangraph_map.pt", "runtime2pertgraph_map.pt",
              "entry2runtimes.joblib", "processed_df.csv", "processed_resource_df.csv",
              "tr2ts_map.joblib"):
        assert (pdir / f).exists(), f
    # tr2data schema
    k, v = next(iter(tr2data.items()))
    assert set(v.keys()) == {"entry_id", "runtime_id", "timestamp", "y"}
    assert torch.is_tensor(v["y"]) and v["y"].dim() == 0
    # graph map schema
    g = next(iter(runtime2span.values()))
    assert set(g.keys()) == {"edge_index", "ms_id", "occurences", "num_nodes", "node_depth", "edge_attr"}
    assert g["edge_index"].dtype == torch.long and g["edge_index"].shape[0] == 2
    assert g["edge_attr"].shape[1] == 2
    assert g["ms_id"].shape[1] == 1
    assert g["node_depth"].dtype == torch.long
    gp = next(iter(runtime2pert.values()))
    assert gp["edge_attr"].shape[1] == 4
    # probabilities normalized
    for entry, rts in entry2runtimes.items():
        assert abs(sum(rts.values()) - 1.0) < 1e-9
    # reload from disk matches in-memory
    on_disk = torch.load(pdir / "tr2data.pt", weights_only=False)
    assert set(on_disk.keys()) == set(tr2data.keys())
    e2r = joblib.load(pdir / "entry2runtimes.joblib")
    assert e2r.keys() == entry2runtimes.keys()


def test_pert_graph_structure(synthetic_workspace):
    _, (_, _, _, runtime2pert, _) = synthetic_workspace
    for g in runtime2pert.values():
        ei = g["edge_index"]
        ea = g["edge_attr"]
        n = g["num_nodes"]
        assert int(ei.max()) + 1 == n
        assert ei.shape[1] == ea.shape[0]
        # intra-ms chain edges carry attr [0,0,1,1] (misc.py:247)
        intra = (ea[:, 2] == 1) & (ea[:, 3] == 1)
        assert (ea[intra][:, :2] == 0).all()
        # call + return edges have same-ms indicator 0
        assert (ea[~intra][:, 3] == 0).all()
        # ms_id length == num stage nodes
        assert g["ms_id"].shape[0] == n


def test_data_list_and_collate(synthetic_workspace):
    root, (tr2data, entry2runtimes, _, runtime2pert, resource_df) = synthetic_workspace
    data_list = build_data_list(tr2data, entry2runtimes, runtime2pert, resource_df)
    assert len(data_list) == len(tr2data)
    s = data_list[0]
    assert s.x.shape[1] == 9
    assert s.x.shape[0] == s.cat_X.shape[0] == s.rt_probs.shape[0] == s.pattern_num_nodes.shape[0]
    assert s.edge_attr.shape[0] == s.edge_index.shape[1]
    # per-node probs: within each pattern, constant and equal to pattern prob
    probs = entry2runtimes[int(s.entry_id)]
    assert abs(float(s.rt_probs.sum()) - sum(
        p * runtime2pert[r]["num_nodes"] for r, p in probs.items())) < 1e-4

    b = collate(data_list[:5])
    assert b.num_graphs == 5
    assert b.x.shape[0] == sum(d.num_nodes for d in data_list[:5])
    assert b.edge_index.shape[1] == sum(d.num_edges for d in data_list[:5])
    # edges are dst-sorted; CSR is consistent
    dst = b.edge_index[1]
    assert (dst[1:] >= dst[:-1]).all()
    n = b.x.shape[0]
    for i in [0, n // 2, n - 1]:
        lo, hi = int(b.row_ptr[i]), int(b.row_ptr[i + 1])
        assert (dst[lo:hi] == i).all()
    # CSC consistency: csc_eid maps to edges whose src is sorted
    src_sorted = b.edge_index[0][b.csc_eid.long()]
    assert (src_sorted[1:] >= src_sorted[:-1]).all()
    # batch vector boundaries
    assert int(b.batch.max()) + 1 == 5


def test_quirk5_duplicate_ms_features(synthetic_workspace):
    """Only the LAST stage node of a duplicated ms carries features."""
    root, (tr2data, entry2runtimes, _, runtime2pert, resource_df) = synthetic_workspace
    data_list = build_data_list(tr2data, entry2runtimes, runtime2pert, resource_df, limit=20)
    found_dup = False
    for s in data_list:
        ms = s.cat_X.flatten().tolist()
        # find a duplicated ms whose feature row is non-missing somewhere
        from collections import Counter
        cnt = Counter(ms)
        for m, c in cnt.items():
            if c > 1:
                nids = [i for i, v in enumerate(ms) if v == m]
                present = [i for i in nids if s.x[i, -1] == 0]
                if present:
                    found_dup = True
                    # only one stage node (the last within its pattern block)
                    # carries features; the rest are zero+missing
                    for i in nids:
                        if i not in present:
                            assert s.x[i, :-1].abs().sum() == 0
                            assert s.x[i, -1] == 1
    assert found_dup, "synthetic data should exercise the duplicate-ms path"


def test_end_to_end_training_epoch(synthetic_workspace):
    root, (tr2data, entry2runtimes, _, runtime2pert, resource_df) = synthetic_workspace
    data_list = build_data_list(tr2data, entry2runtimes, runtime2pert, resource_df)
    train, valid, test_ = split_60_20_20(data_list)
    assert len(train) + len(valid) + len(test_) == len(data_list)

    cat_max = max(int(s.cat_X.max()) for s in data_list)
    entry_max = max(int(s.entry_id) for s in data_list)
    ifc_max = max(int(s.edge_attr[:, 0].max()) for s in data_list)
    rpc_max = max(int(s.edge_attr[:, 1].max()) for s in data_list)
    model = SAGEDeterministic(9, [cat_max + 1], entry_max, ifc_max, rpc_max,
                              hidden_channels=16, num_layers=1, dropout=0.0)
    opt = torch.optim.Adam(model.parameters(), lr=3e-3)
    loader = BatchLoader(train, batch_size=16, shuffle=True, seed=0)
    model.train()
    epoch_means = []
    for epoch in range(3):
        losses = []
        for b in loader:
            opt.zero_grad()
            gp, lp = model(b.x, b.cat_X, b.edge_index, b.edge_attr,
                           b.pattern_num_nodes, b.rt_probs, b.entry_id, b.batch,
                           csr=b.csr, num_graphs=b.num_graphs)
            loss = F.quantile_loss(b.y, gp.flatten(), 0.5)
            loss.backward()
            opt.step()
            losses.append(float(loss.detach()))
        epoch_means.append(sum(losses) / len(losses))
    assert epoch_means[-1] < epoch_means[0]


def test_native_collator_matches_python(synthetic_workspace):
    """csrc/collate.cpp vs the torch collator — identical batch content."""
    from pertgnn.ops.backend import ext
    import pytest as _pytest
    if ext() is None:
        _pytest.skip("extension not built")
    from pertgnn.data.collate import collate_native
    root, (tr2data, entry2runtimes, _, runtime2pert, resource_df) = synthetic_workspace
    data_list = build_data_list(tr2data, entry2runtimes, runtime2pert, resource_df, limit=40)
    py = collate(data_list[:12])
    nat = collate_native(data_list[:12])
    assert torch.equal(py.x, nat.x)
    assert torch.equal(py.cat_X, nat.cat_X)
    assert torch.equal(py.edge_index, nat.edge_index)
    assert torch.equal(py.edge_attr, nat.edge_attr)
    assert torch.equal(py.rt_probs, nat.rt_probs)
    assert torch.equal(py.batch, nat.batch)
    assert torch.equal(py.y, nat.y)
    assert torch.equal(py.row_ptr, nat.row_ptr)
    assert torch.equal(py.csr_src, nat.csr_src)
    assert torch.equal(py.col_ptr, nat.col_ptr)
    assert torch.equal(py.csc_dst, nat.csc_dst)
    assert torch.equal(py.csc_eid, nat.csc_eid)
    assert torch.equal(nat.batch_ptr[1:].long(), torch.cumsum(
        torch.bincount(py.batch, minlength=py.num_graphs), 0).int().long())


def test_checkpoint_resume_cli(synthetic_workspace, tmp_path):
    """save_checkpoint/load_checkpoint round-trip with FusedAdam state."""
    from pertgnn.train import load_checkpoint, save_checkpoint
    from pertgnn.train.optim import FusedAdam

    root, (tr2data, entry2runtimes, _, runtime2pert, resource_df) = synthetic_workspace
    data_list = build_data_list(tr2data, entry2runtimes, runtime2pert, resource_df, limit=30)
    model = SAGEDeterministic(9, [64], 8, 50, 8, hidden_channels=16, num_layers=1, dropout=0.0)
    opt = FusedAdam(model.parameters(), lr=1e-3)
    loader = BatchLoader(data_list, batch_size=8, shuffle=False)
    model.train()
    for b in loader:
        opt.zero_grad()
        gp, _ = model(b.x, b.cat_X, b.edge_index, b.edge_attr,
                      b.pattern_num_nodes, b.rt_probs, b.entry_id, b.batch,
                      csr=b.csr, num_graphs=b.num_graphs)
        F.quantile_loss(b.y, gp.flatten(), 0.5).backward()
        opt.step()
        break
    p = tmp_path / "ck.pt"
    save_checkpoint(str(p), model, opt, epoch=3, extra={"note": "x"})

    model2 = SAGEDeterministic(9, [64], 8, 50, 8, hidden_channels=16, num_layers=1, dropout=0.0)
    opt2 = FusedAdam(model2.parameters(), lr=1e-3)
    epoch, extra = load_checkpoint(str(p), model2, opt2)
    assert epoch == 3 and extra["note"] == "x"
    assert opt2.step_count == opt.step_count
    for (n1, p1), (n2, p2) in zip(model.named_parameters(), model2.named_parameters()):
        if isinstance(p1, torch.nn.parameter.UninitializedParameter):
            continue
        assert torch.equal(p1.detach(), p2.detach()), n1
    assert torch.equal(opt.exp_avg, opt2.exp_avg)


def test_cli_end_to_end_subprocess(tmp_path):
    """The real `python pert_gnn.py` entrypoint: synthetic data, 2 epochs on
    CPU with precision/loss_scale flags — catches argument-wiring and
    module-level regressions no unit test sees."""
    import json
    import re
    import subprocess
    import sys

    metrics = tmp_path / "m.jsonl"
    cmd = [sys.executable, "pert_gnn.py", "--synthetic", "--graph_type", "pert",
           "--epochs", "2", "--num_layers", "1", "--hidden_channels", "16",
           "--batch_size", "32", "--seed", "3", "--loss_scale", "8",
           "--processed_dir", str(tmp_path / "processed"),
           "--metrics_jsonl", str(metrics)]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                       cwd=str(Path(__file__).resolve().parents[1]))
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("Epoch:")]
    assert len(lines) == 2, r.stdout[-2000:]
    # reference epoch line format
    assert re.match(r"Epoch: 2, Train: [\d.]+, Test mae: [\d.]+, ", lines[1])
    recs = [json.loads(l) for l in metrics.read_text().splitlines()]
    assert len(recs) == 2 and recs[1]["epoch"] == 2
    assert all("test_mae" in r or "test" in str(r) for r in recs)


def test_cli_hipgraph_flag_cpu(tmp_path):
    """--hipgraph on CPU: GraphStepper degrades to eager stepping over the
    fixed-composition resident batches; the CLI must still train and emit
    the reference epoch lines."""
    import subprocess
    import sys

    cmd = [sys.executable, "pert_gnn.py", "--synthetic", "--graph_type", "pert",
           "--epochs", "2", "--num_layers", "1", "--hidden_channels", "16",
           "--batch_size", "32", "--seed", "3", "--hipgraph",
           "--processed_dir", str(tmp_path / "processed")]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                       cwd=str(Path(__file__).resolve().parents[1]))
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("Epoch:")]
    assert len(lines) == 2, r.stdout[-2000:]
    assert any("hipgraph stepping mode: eager" in l for l in r.stdout.splitlines())


def test_graph_stepper_semantics_cpu(synthetic_workspace):
    """GraphStepper (eager mode) must train exactly like a manual loop over
    the same fixed batches in the same order — including the capture()
    snapshot/restore noop on CPU."""
    from pertgnn.models import SAGEDeterministic
    from pertgnn.ops import functional as F
    from pertgnn.train.capture import GraphStepper, make_resident_batches
    from pertgnn.train.optim import FusedAdam
    from pertgnn.data.collate import collate_native

    root, (tr2data, entry2runtimes, _, runtime2pert, resource_df) = synthetic_workspace
    data_list = build_data_list(tr2data, entry2runtimes, runtime2pert, resource_df, limit=48)
    batches = make_resident_batches(data_list, 16, None, seed=5,
                                    collate_fn=collate_native)

    def build():
        torch.manual_seed(9)
        m = SAGEDeterministic(9, [64], 50, 50, 10, 16, 1, 0.0)
        return m, FusedAdam(m.parameters(), lr=1e-3)

    m1, o1 = build()
    stepper = GraphStepper(m1, o1, None, None, 0.5, torch.device("cpu"),
                           batches, seed=5)
    assert stepper.capture() == "eager"
    loss_sum, mape_sum, n = stepper.run_epoch_sums()
    assert n == sum(b.num_graphs for b in batches)

    m2, o2 = build()
    order = torch.randperm(len(batches),
                           generator=torch.Generator().manual_seed(5)).tolist()
    tot = 0.0
    for i in order:
        b = batches[i]
        o2.zero_grad(set_to_none=False)
        gp, _ = m2(b.x, b.cat_X, b.edge_index, b.edge_attr,
                   b.pattern_num_nodes, b.rt_probs, b.entry_id, b.batch,
                   csr=b.csr, num_graphs=b.num_graphs)
        loss = F.quantile_loss(b.y, gp.flatten(), 0.5)
        loss.backward()
        o2.step()
        tot += float(loss.detach()) * b.num_graphs
    assert torch.equal(o1.flat_param, o2.flat_param)
    assert abs(tot - loss_sum) < 1e-3 * max(abs(tot), 1.0)


def test_threaded_loader_equivalence(synthetic_workspace):
    """ThreadedLoader yields the same batches as iterating the base loader."""
    from pertgnn.data.prefetch import ThreadedLoader

    root, (tr2data, entry2runtimes, _, runtime2pert, resource_df) = synthetic_workspace
    data_list = build_data_list(tr2data, entry2runtimes, runtime2pert, resource_df, limit=24)
    base = BatchLoader(data_list, batch_size=8, shuffle=False)
    ref_batches = list(base)
    thr_batches = list(ThreadedLoader(BatchLoader(data_list, batch_size=8, shuffle=False)))
    assert len(ref_batches) == len(thr_batches)
    for a, b in zip(ref_batches, thr_batches):
        assert torch.equal(a.x, b.x)
        assert torch.equal(a.edge_index, b.edge_index)
        assert torch.equal(a.y, b.y)


def test_ingest_cached_csv_reload(synthetic_workspace, tmp_path):
    """get_df's second call reads the cached processed CSVs and reproduces
    the same frame (preprocess.py:191-266 cache contract)."""
    import pandas as pd

    from pertgnn.data.ingest import get_df

    root, _ = synthetic_workspace
    processed = os.path.join(root, "processed")
    df1, rs1 = get_df(os.path.join(root, "data"), processed)
    assert os.path.isfile(os.path.join(processed, "processed_df.csv"))
    df2, rs2 = get_df(os.path.join(root, "data"), processed)
    pd.testing.assert_frame_equal(
        df1.reset_index(drop=True), df2.reset_index(drop=True), check_dtype=False)
    pd.testing.assert_frame_equal(
        rs1.reset_index(drop=True), rs2.reset_index(drop=True), check_dtype=False)


def test_reference_pyg_cache_loads_without_pyg(tmp_path):
    """full_*_data_list.pt written by the REFERENCE is a pickled list of
    PyG Data objects (reference pert_gnn.py:317-322).  The shim unpickler
    loads it without torch_geometric installed and converts to TraceSample
    (rt_probs derived from the contiguous-pattern layout)."""
    import sys
    import types

    # fabricate a minimal torch_geometric so we can WRITE a PyG-shaped
    # pickle, then remove it so the LOAD runs without PyG (like this env)
    tg = types.ModuleType("torch_geometric")
    tg_data = types.ModuleType("torch_geometric.data")
    tg_data_data = types.ModuleType("torch_geometric.data.data")

    class GlobalStorage:
        def __init__(self, mapping):
            self._mapping = mapping

    class Data:
        def __init__(self, **kw):
            self._store = GlobalStorage(dict(kw))

    for cls in (GlobalStorage, Data):
        cls.__module__ = "torch_geometric.data.data"
        cls.__qualname__ = cls.__name__
    tg_data_data.Data = Data
    tg_data_data.GlobalStorage = GlobalStorage
    sys.modules["torch_geometric"] = tg
    sys.modules["torch_geometric.data"] = tg_data
    sys.modules["torch_geometric.data.data"] = tg_data_data
    try:
        # two patterns: 3 nodes then 2 nodes
        d = Data(
            x=torch.randn(5, 9),
            edge_index=torch.tensor([[0, 1, 3], [1, 2, 4]]),
            edge_attr=torch.tensor([[1, 0], [0, 1], [2, 1]]),
            cat_X=torch.tensor([[0], [1], [2], [0], [1]]),
            node_depth=torch.tensor([[0], [1], [2], [0], [1]]),
            pattern_num_nodes=torch.tensor([[3.], [3.], [3.], [2.], [2.]]),
            pattern_probs=torch.tensor([[0.75], [0.25]]),
            entry_id=torch.tensor([4]),
            y=torch.tensor(42.0),
        )
        path = tmp_path / "full_pert_data_list.pt"
        torch.save([d, d], str(path))
    finally:
        for m in ("torch_geometric.data.data", "torch_geometric.data",
                  "torch_geometric"):
            del sys.modules[m]

    from pertgnn.data.pyg_compat import load_data_list_any

    lst = load_data_list_any(str(path))
    assert len(lst) == 2
    s = lst[0]
    assert s.x.shape == (5, 9) and float(s.y) == 42.0
    assert int(s.entry_id) == 4
    # rt_probs expanded per node from the contiguous pattern layout
    assert torch.allclose(s.rt_probs.flatten(),
                          torch.tensor([0.75, 0.75, 0.75, 0.25, 0.25]))
    # our own format still loads through the same entry point
    torch.save(lst, str(path))
    lst2 = load_data_list_any(str(path))
    assert torch.equal(lst2[1].x, s.x)


def test_bench_json_contract_cpu(tmp_path):
    """The driver parses ONE JSON line from bench.py rank 0 — guard the
    contract keys and their semantics (CPU run, tiny config)."""
    import json
    import subprocess
    import sys

    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--batch-size", "16", "--n-batches", "1", "--layers", "2",
         "--hidden", "32", "--vocab", "small", "--mae-epochs", "0"],
        capture_output=True, text=True, timeout=420,
        cwd=str(Path(__file__).resolve().parents[1]),
    )
    assert out.returncode == 0, out.stderr[-2000:]
    rec = json.loads(out.stdout.strip().splitlines()[-1])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config", "latency_mae"):
        assert key in rec, key
    assert rec["n_gpus"] == 1 and rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["scaling"] == "weak" and rec["higher_is_better"] is True
    assert rec["data"] == "synthetic"
    assert rec["config"]["global_batch"] == 16
    assert rec["value"] > 0 and rec["ms_per_step"] > 0


def test_cli_dynamic_loss_scale_cpu(tmp_path):
    """--loss_scale dynamic parses and trains on CPU (eager GradScaler
    semantics in FusedAdam)."""
    import subprocess
    import sys

    cmd = [sys.executable, "pert_gnn.py", "--synthetic", "--graph_type", "pert",
           "--epochs", "2", "--num_layers", "1", "--hidden_channels", "16",
           "--batch_size", "32", "--seed", "3", "--loss_scale", "dynamic",
           "--processed_dir", str(tmp_path / "processed")]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                       cwd=str(Path(__file__).resolve().parents[1]))
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("Epoch:")]
    assert len(lines) == 2


def test_reference_pyg_cache_flat_layout(tmp_path):
    """Older PyG pickles keep tensors as direct Data attributes (no
    _store/_mapping nesting) — the shim's tensor walk must find both
    layouts (pertgnn/data/pyg_compat.py:_find_tensors)."""
    import sys
    import types

    tg = types.ModuleType("torch_geometric")
    tg_data = types.ModuleType("torch_geometric.data")

    class Data:
        def __init__(self, **kw):
            self.__dict__.update(kw)

    Data.__module__ = "torch_geometric.data"
    Data.__qualname__ = "Data"
    tg_data.Data = Data
    sys.modules["torch_geometric"] = tg
    sys.modules["torch_geometric.data"] = tg_data
    try:
        d = Data(
            x=torch.randn(3, 9),
            edge_index=torch.tensor([[0, 1], [1, 2]]),
            edge_attr=torch.tensor([[1, 0], [0, 1]]),
            cat_X=torch.tensor([[0], [1], [2]]),
            node_depth=torch.tensor([[0], [1], [2]]),
            pattern_num_nodes=torch.tensor([[3.], [3.], [3.]]),
            pattern_probs=torch.tensor([[1.0]]),
            entry_id=torch.tensor([7]),
            y=torch.tensor(5.0),
        )
        path = tmp_path / "full_span_data_list.pt"
        torch.save([d], str(path))
    finally:
        del sys.modules["torch_geometric.data"], sys.modules["torch_geometric"]

    from pertgnn.data.pyg_compat import load_data_list_any

    lst = load_data_list_any(str(path))
    assert len(lst) == 1 and int(lst[0].entry_id) == 7
    assert torch.allclose(lst[0].rt_probs.flatten(), torch.ones(3))
