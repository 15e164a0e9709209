import sys, os
sys.path.insert(0, '/root/repo')
import torch

det = os.environ.get("D", "0") == "1"
if det:
    os.environ["PERTGNN_DETERMINISTIC"] = "1"
vocab = int(os.environ.get("V", "40"))

from pertgnn.ops import functional as F
from pertgnn.ops.backend import ext
from pertgnn.data.collate import build_csr

DEV = torch.device("cuda:0")
m = ext()

def ck(tag):
    torch.cuda.synchronize()
    print("OK", tag, flush=True)

torch.manual_seed(0)
n, e, h = 5300, 7000, 256
src = torch.randint(0, n, (e,))
dst = torch.randint(0, n, (e,))
ei = torch.stack([src, dst])
perm, row_ptr, csr_src, col_ptr, csc_dst, csc_eid = build_csr(ei, n)
csr = tuple(t.to(DEV) for t in (row_ptr, csr_src, col_ptr, csc_dst, csc_eid))
ea = torch.stack([torch.randint(0, vocab, (e,)),
                  torch.randint(0, 5, (e,))], 1).to(DEV)
qkvs = (torch.randn(n, 4 * h) * 0.3).to(torch.bfloat16).to(DEV).requires_grad_(True)
ifc_w = torch.randn(vocab, h, device=DEV)
rpc_w = torch.randn(5, h, device=DEV)
we = torch.randn(h, h, device=DEV) * 0.05

pifc = F.linear16(ifc_w, we)
prpc = F.linear16(rpc_w, we)
ck("linear16 P")
pifc = pifc.detach().requires_grad_(True)
prpc = prpc.detach().requires_grad_(True)
out = F.edge_attention_fused(qkvs, pifc, prpc, ea, csr, out16=True)
ck("fwd p16")
out.float().pow(2).sum().backward()
ck("bwd p16")
print("grads", pifc.grad.dtype, float(qkvs.grad.float().abs().sum()))
# direct table grad on bf16 de
de = (torch.randn(e, h) * 0.1).to(torch.bfloat16).to(DEV)
dt = F._table_grad(m, de, ea[:, 0], vocab + 2, h, 0)
ck("table_grad bf16")
print("DONE")
