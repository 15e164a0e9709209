"""Reader-side compatibility for the reference's `full_*_data_list.pt` caches.

The reference caches its assembled dataset as a pickled list of PyG
``Data`` objects (reference pert_gnn.py:317-322).  This framework has no
PyG dependency, so those pickles normally fail to load (unpickling needs
the ``torch_geometric`` classes).  ``load_data_list_any`` accepts BOTH
formats:

  * our own caches (a list of TraceSample) load directly;
  * reference caches load through a shim Unpickler that maps every
    ``torch_geometric.*`` class to a generic state-absorbing stand-in, then
    extracts the tensor fields and converts each object to a TraceSample
    (``rt_probs`` — which the reference rebuilds per batch from its
    lru-cached helpers, pert_gnn.py:220-230 — is derived from
    pattern_probs × pattern_num_nodes, same values).

Writing stays in TraceSample format (documented deviation, README): the
tensor fields and names match the reference Data exactly, only the
container class differs.
"""
from __future__ import annotations

import pickle

import torch


class _ShimBase:
    """Absorbs any pickled state: works for PyG Data / BaseStorage /
    GlobalStorage and friends across PyG versions (their __reduce__ carries
    plain dicts once weakrefs are stripped for pickling)."""

    def __init__(self, *args, **kwargs):
        self._shim_args = args
        self.__dict__.update(kwargs)

    def __setstate__(self, state):
        if isinstance(state, dict):
            self.__dict__.update(state)
        else:
            self._shim_state = state


class _ShimUnpickler(pickle.Unpickler):
    # no persistent_load here: torch.load assigns its own onto the instance
    # (defining one would get cached by the C unpickler and shadow torch's)
    def find_class(self, module, name):
        if module.startswith("torch_geometric"):
            return type(f"Shim_{name}", (_ShimBase,), {})
        return super().find_class(module, name)


_FIELDS = ("x", "edge_index", "edge_attr", "cat_X", "node_depth",
           "pattern_num_nodes", "pattern_probs", "entry_id", "y")


def _find_tensors(obj, depth=0):
    """Walk a shim object graph for a dict holding the Data tensor fields
    (PyG keeps them in Data._store._mapping; be tolerant of layout)."""
    if depth > 6:
        return None
    if isinstance(obj, dict):
        if "x" in obj and "edge_index" in obj:
            return obj
        for v in obj.values():
            got = _find_tensors(v, depth + 1)
            if got is not None:
                return got
        return None
    if isinstance(obj, _ShimBase):
        return _find_tensors(obj.__dict__, depth + 1)
    return None


def _to_sample(obj):
    from .dataset import TraceSample

    if isinstance(obj, TraceSample):
        return obj
    mapping = _find_tensors(obj)
    if mapping is None:
        raise ValueError(f"cannot extract Data fields from {type(obj).__name__}")
    fields = {k: mapping[k] for k in _FIELDS if k in mapping}
    missing = [k for k in _FIELDS if k not in fields]
    if missing:
        raise ValueError(f"reference cache object lacks fields {missing}")
    # rt_probs: expand per-pattern probs to per-node (patterns are stored
    # contiguously; pattern_num_nodes[i] = size of node i's pattern)
    pnn = fields["pattern_num_nodes"].reshape(-1)
    probs = fields["pattern_probs"].reshape(-1)
    n = pnn.numel()
    rt = torch.empty(n, 1, dtype=torch.float32)
    pos = 0
    p = 0
    while pos < n:
        cnt = int(pnn[pos])
        cnt = max(cnt, 1)
        rt[pos:pos + cnt, 0] = float(probs[p]) if p < probs.numel() else 0.0
        pos += cnt
        p += 1
    entry_id = fields["entry_id"].reshape(-1)[:1].to(torch.long)
    return TraceSample(
        x=fields["x"].float(),
        edge_index=fields["edge_index"].long(),
        edge_attr=fields["edge_attr"].long(),
        cat_X=fields["cat_X"].long(),
        node_depth=fields["node_depth"],
        pattern_num_nodes=fields["pattern_num_nodes"].float(),
        pattern_probs=fields["pattern_probs"].float(),
        rt_probs=rt,
        entry_id=entry_id,
        y=fields["y"].reshape(()).float(),
    )


def load_data_list_any(path):
    """Load a data-list cache written by THIS framework (TraceSample list)
    or by the REFERENCE (PyG Data list) — the latter without PyG installed."""
    try:
        data_list = torch.load(path, weights_only=False)
    except ModuleNotFoundError as exc:
        if "torch_geometric" not in str(exc):
            raise
        data_list = torch.load(path, weights_only=False,
                               pickle_module=_shim_pickle_module())
    if not isinstance(data_list, (list, tuple)):
        raise ValueError(f"{path} does not contain a data list")
    from .dataset import TraceSample

    if all(isinstance(s, TraceSample) for s in data_list):
        return list(data_list)
    return [_to_sample(s) for s in data_list]


def _shim_pickle_module():
    """A pickle-module stand-in whose Unpickler shims torch_geometric
    classes (torch.load instantiates ``pickle_module.Unpickler``)."""
    import types

    mod = types.ModuleType("pertgnn_shim_pickle")
    mod.Unpickler = _ShimUnpickler
    mod.load = pickle.load
    mod.loads = pickle.loads
    return mod
