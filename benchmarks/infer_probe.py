import sys, time, torch
sys.path.insert(0, ".")
import bench as bench_mod
from pertgnn.models import SAGEDeterministic
from pertgnn.ops.functional import set_gemm_precision

dev = torch.device("cuda:0")
set_gemm_precision("bf16")
batches, stats = bench_mod.build_synthetic_batches(2, 1024, seed=0, device=dev)
model = SAGEDeterministic(9, [stats["cat_max"] + 1], stats["entry_max"],
                          stats["ifc_max"], stats["rpc_max"], 256, 8, 0.0).to(dev)
model.eval()
with torch.no_grad():
    for b in batches:
        model(b.x, b.cat_X, b.edge_index, b.edge_attr, b.pattern_num_nodes,
              b.rt_probs, b.entry_id, b.batch, csr=b.csr, num_graphs=b.num_graphs)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 30
    for i in range(iters):
        b = batches[i % 2]
        model(b.x, b.cat_X, b.edge_index, b.edge_attr, b.pattern_num_nodes,
              b.rt_probs, b.entry_id, b.batch, csr=b.csr, num_graphs=b.num_graphs)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
print(f"inference: {1024*iters/dt:.0f} graphs/s ({dt/iters*1e3:.2f} ms / 1024-graph batch, eager fwd)")
