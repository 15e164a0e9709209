import sys, os
sys.path.insert(0, '/root/repo')
if os.environ.get("D", "1") == "1":
    os.environ["PERTGNN_DETERMINISTIC"] = "1"
import torch
import bench as bench_mod
from pertgnn.models import SAGEDeterministic
from pertgnn.ops import functional as F
from pertgnn.ops.functional import set_gemm_precision

DEV = torch.device("cuda:0")
torch.manual_seed(2)
batches, stats = bench_mod.build_synthetic_batches(1, 32, seed=7, device=DEV)
b = batches[0]
model = SAGEDeterministic(9, [stats["cat_max"] + 1], stats["entry_max"],
                          stats["ifc_max"], stats["rpc_max"], 256, int(os.environ.get("L", "3")), 0.0).to(DEV)
model.train()
set_gemm_precision("bf16")

def grads():
    gp, _ = model(b.x, b.cat_X, b.edge_index, b.edge_attr,
                  b.pattern_num_nodes, b.rt_probs, b.entry_id, b.batch,
                  csr=b.csr, num_graphs=b.num_graphs)
    loss = F.quantile_loss(b.y, gp.flatten(), 0.5)
    loss.backward()
    out = {n: p.grad.clone() for n, p in model.named_parameters()
           if p.grad is not None}
    model.zero_grad()
    torch.cuda.synchronize()
    return out

g1 = grads()
g2 = grads()
bad = [(n, (g1[n].float() - g2[n].float()).abs().max().item())
       for n in g1 if not torch.equal(g1[n], g2[n])]
print("MISMATCH:", bad if bad else "none", flush=True)
