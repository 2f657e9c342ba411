import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from metaflow_amd.ops import kernels as K

def rel(a, b):
    a, b = a.float(), b.float()
    return ((a - b).norm() / (b.norm() + 1e-8)).item()

ext = K.hip_ext()
dev = "cuda:0"
torch.manual_seed(0)
kmat = torch.randn(64, 128, dtype=torch.bfloat16, device=dev)
qmat = torch.randn(32, 128, dtype=torch.bfloat16, device=dev)
st = ext.dbg_st(kmat, qmat)
st_ref = kmat.float() @ qmat.float().T
print("dbg_st err:", rel(st, st_ref))

pmat = torch.randn(64, 32, dtype=torch.bfloat16, device=dev)
bmat = torch.randn(32, 128, dtype=torch.bfloat16, device=dev)
dv = ext.dbg_dv(pmat, bmat)
dv_ref = pmat.float() @ bmat.float()
print("dbg_dv err:", rel(dv, dv_ref))
# where is it wrong?
e = (dv - dv_ref).view(4, 16, 8, 16)
r = dv_ref.view(4, 16, 8, 16)
print("dv err by (wave strip, n-tile):")
for w in range(4):
    print("  w%d:" % w, " ".join("%.2f" % rel(e[w,:,n,:] + r[w,:,n,:], r[w,:,n,:]) for n in range(8)))
