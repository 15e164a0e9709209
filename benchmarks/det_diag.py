import sys
sys.path.insert(0, '/root/repo')
import os, torch
os.environ["PERTGNN_DETERMINISTIC"] = "1"
import bench as bench_mod
from pertgnn.models import SAGEDeterministic
from pertgnn.ops import functional as F
from pertgnn.ops.functional import set_gemm_precision

DEV = torch.device("cuda:0")
torch.manual_seed(2)
batches, stats = bench_mod.build_synthetic_batches(1, 32, seed=7, device=DEV)
b = batches[0]
model = SAGEDeterministic(9, [stats["cat_max"] + 1], stats["entry_max"],
                          stats["ifc_max"], stats["rpc_max"], 256, 3, 0.0).to(DEV)
model.train()
set_gemm_precision("bf16")

acts = {}
def run(tag):
    rec = {}
    hooks = []
    def mk(name):
        def h(mod, inp, out):
            pass
        return h
    # capture intermediates manually by monkeypatching ops
    import pertgnn.ops.functional as fn
    orig_att = fn.edge_attention_fused
    orig_lin16 = fn.linear16
    calls = {"att": 0, "lin16": 0}
    def att(*a, **k):
        out = orig_att(*a, **k)
        rec[f"att{calls['att']}"] = out.detach().clone()
        rec[f"qkvs{calls['att']}"] = a[0].detach().clone()
        rec[f"pifc{calls['att']}"] = a[1].detach().clone()
        calls["att"] += 1
        return out
    def lin16(*a, **k):
        out = orig_lin16(*a, **k)
        rec[f"lin16_{calls['lin16']}"] = out.detach().clone()
        calls["lin16"] += 1
        return out
    fn.edge_attention_fused = att
    fn.linear16 = lin16
    import pertgnn.models.pert_gnn as mp
    orig_ops_att = mp.ops.edge_attention_fused
    mp.ops.edge_attention_fused = att
    orig_ops_lin = mp.ops.linear16
    mp.ops.linear16 = lin16
    try:
        gp, _ = model(b.x, b.cat_X, b.edge_index, b.edge_attr,
                      b.pattern_num_nodes, b.rt_probs, b.entry_id, b.batch,
                      csr=b.csr, num_graphs=b.num_graphs)
        loss = F.quantile_loss(b.y, gp.flatten(), 0.5)
        loss.backward()
        rec["gp"] = gp.detach().clone()
        for n, p in model.named_parameters():
            if p.grad is not None:
                rec["grad:" + n] = p.grad.clone()
        model.zero_grad()
    finally:
        fn.edge_attention_fused = orig_att
        fn.linear16 = orig_lin16
        mp.ops.edge_attention_fused = orig_ops_att
        mp.ops.linear16 = orig_ops_lin
    return rec

r1 = run("a")
r2 = run("b")
bad = []
for k in r1:
    if not torch.equal(r1[k], r2[k]):
        bad.append((k, (r1[k].float() - r2[k].float()).abs().max().item()))
print("MISMATCHES:", bad if bad else "none")
