"""Diagnostic: log the (m,n,k) of every linear16 GEMM in one flagship step
and which routing gate each wgrad/dgrad call hits (run on a GPU box:
`PYTHONPATH=. python tools/diag_wgrad_shapes.py`)."""
import collections
import torch

import pertgnn.ops.backend as B

real_ext = B.ext
calls = collections.Counter()

LOG = {"linear_wgrad16_b16", "linear_wgrad16", "linear_dgrad16_o16",
       "linear_dgrad16", "linear_fwd_a16o16", "linear_fwd_bf16_o16"}


class Proxy:
    def __init__(self, m):
        self._m = m

    def __getattr__(self, name):
        fn = getattr(self._m, name)
        if name not in LOG:
            return fn

        def wrap(*a, **k):
            if name.startswith("linear_wgrad"):
                g, x = a[0], a[1]
                m_, k_, n_ = x.shape[0], x.shape[1], g.shape[1]
                gate = ("glds" if m_ >= 512 and n_ % 128 == 0 and k_ % 128 == 0
                        else "fallback")
            elif name.startswith("linear_dgrad"):
                g, w = a[0], a[1]
                m_, n_, k_ = g.shape[0], g.shape[1], w.shape[1]
                gate = ("glds" if k_ % 128 == 0 and n_ % 64 == 0 and m_ > 16
                        else ("skinny" if m_ <= 16 else "fallback"))
            else:
                x, w = a[0], a[1]
                m_, k_, n_ = x.shape[0], x.shape[1], w.shape[0]
                gate = ("glds" if m_ >= 512 and n_ % 128 == 0 and k_ % 32 == 0
                        else "fallback")
            calls[(name, m_, n_, k_, gate)] += 1
            return fn(*a, **k)
        return wrap


B.ext = lambda: Proxy(real_ext())
import pertgnn.ops.functional as F
F.ext = B.ext

import sys
sys.argv = ["bench"]
from bench import build_synthetic_batches
from pertgnn.models import SAGEDeterministic
from pertgnn.ops import functional as OF
from pertgnn.ops.functional import set_gemm_precision

dev = torch.device("cuda")
set_gemm_precision("bf16")
batches, stats = build_synthetic_batches(1, 1024, seed=100, device=dev,
                                         vocab="realistic")
model = SAGEDeterministic(9, [stats["cat_max"] + 1], stats["entry_max"],
                          stats["ifc_max"], stats["rpc_max"], 256, 8, 0.0).to(dev)
b = batches[0]
model.train()
gp, _ = model(b.x, b.cat_X, b.edge_index, b.edge_attr, b.pattern_num_nodes,
              b.rt_probs, b.entry_id, b.batch, csr=b.csr,
              num_graphs=b.num_graphs)
loss = OF.quantile_loss(b.y, gp.flatten(), 0.5)
loss.backward()
torch.cuda.synchronize()
for (name, m_, n_, k_, gate), c in sorted(calls.items()):
    flops = 2 * m_ * n_ * k_ * c
    print(f"{name:22s} m={m_:7d} n={n_:5d} k={k_:5d} x{c:2d} {gate:8s} "
          f"{flops/1e9:8.2f} GFLOP")
