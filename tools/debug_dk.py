import os, sys, math
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from metaflow_amd.ops import kernels as K

def rel(a, b):
    a, b = a.float(), b.float()
    return ((a - b).norm() / (b.norm() + 1e-8)).item()

ext = K.hip_ext()
dev = "cuda:0"
torch.manual_seed(0)
B, H, Hkv, S = 1, 4, 2, 256
q = torch.randn(B, H, S, 128, dtype=torch.bfloat16, device=dev)
k = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
v = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
scale = 1.0 / math.sqrt(128)
o, lse = ext.attn_fwd(q, k, v, scale)
dout = torch.randn_like(o)
dq, dk, dv = ext.attn_bwd(q, k, v, o, dout, lse, scale)

G = H // Hkv
qf = q.float(); kf = k.float().repeat_interleave(G, 1)
vf = v.float().repeat_interleave(G, 1); df = dout.float(); of = o.float()
s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
mask = torch.ones(S, S, dtype=torch.bool, device=dev).tril()
p = torch.exp(s - lse.unsqueeze(-1)).masked_fill(~mask, 0.0)
delta = (df * of).sum(-1)
dp = torch.matmul(df, vf.transpose(-1, -2))
ds = p * (dp - delta.unsqueeze(-1)) * scale
dq_ref = torch.matmul(ds, kf)
dk_ref = torch.matmul(ds.transpose(-1, -2), qf).view(B, Hkv, G, S, 128).sum(2)
dv_ref = torch.matmul(p.transpose(-1, -2), df).view(B, Hkv, G, S, 128).sum(2)
print("dq %.4f dk %.4f dv %.4f" % (rel(dq, dq_ref), rel(dk, dk_ref), rel(dv, dv_ref)))
def nrm(t, keep):
    dims = tuple(d for d in range(t.dim()) if d != keep)
    return (t*t).sum(dims).sqrt()
e = (dk.float() - dk_ref).view(B*Hkv*(S//32), 32, 8, 16)
r = dk_ref.view(B*Hkv*(S//32), 32, 8, 16)
print("dk err by kvblock:", " ".join("%.2f" % x for x in (nrm(e,0)/(nrm(r,0)+1e-8)).tolist()))
print("dk err by row%32:", " ".join("%.2f" % x for x in (nrm(e,1)/(nrm(r,1)+1e-8)).tolist()))
print("dk err by d16grp:", " ".join("%.2f" % x for x in (nrm(e,2)/(nrm(r,2)+1e-8)).tolist()))
print("dk err by d%16:", " ".join("%.2f" % x for x in (nrm(e,3)/(nrm(r,3)+1e-8)).tolist()))
