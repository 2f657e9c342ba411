"""Debug harness for attention backward: fingerprints which term of
dS = P*(dP - delta)*scale the dq/dk/dv kernels get wrong. GPU-only."""

import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(
    __file__))))

import torch

from metaflow_amd.ops import kernels as K


def rel(a, b):
    a, b = a.float(), b.float()
    return ((a - b).norm() / (b.norm() + 1e-8)).item()


def run_case(B, H, Hkv, S):
    dev = "cuda:0"
    torch.manual_seed(0)
    q = torch.randn(B, H, S, 128, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    scale = 1.0 / math.sqrt(128)
    ext = K.hip_ext()
    o, lse = ext.attn_fwd(q, k, v, scale)
    dout = torch.randn_like(o)
    dq, dk, dv = ext.attn_bwd(q, k, v, o, dout, lse, scale)

    G = H // Hkv
    qf = q.float()
    kf = k.float().repeat_interleave(G, 1)
    vf = v.float().repeat_interleave(G, 1)
    df = dout.float()
    of = o.float()

    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    mask = torch.ones(S, S, dtype=torch.bool, device=dev).tril()
    p = torch.exp(s - lse.unsqueeze(-1)).masked_fill(~mask, 0.0)
    delta = (df * of).sum(-1)                      # [B,H,S]
    dp = torch.matmul(df, vf.transpose(-1, -2))    # [B,H,S,S]
    ds = p * (dp - delta.unsqueeze(-1)) * scale

    dq_ref = torch.matmul(ds, kf)
    dk_full = torch.matmul(ds.transpose(-1, -2), qf)   # per q-head
    dv_full = torch.matmul(p.transpose(-1, -2), df)
    dk_ref = dk_full.view(B, Hkv, G, S, 128).sum(2)
    dv_ref = dv_full.view(B, Hkv, G, S, 128).sum(2)

    print("case B=%d H=%d Hkv=%d S=%d" % (B, H, Hkv, S))
    print("  dq err          %.4f" % rel(dq, dq_ref))
    print("  dk err          %.4f" % rel(dk, dk_ref))
    print("  dv err          %.4f" % rel(dv, dv_ref))
    # fingerprints
    ds_nodelta = p * dp * scale
    ds_onlydelta = p * (-delta.unsqueeze(-1)) * scale
    print("  dq vs nodelta   %.4f" % rel(dq, torch.matmul(ds_nodelta, kf)))
    print("  dq vs onlydelta %.4f"
          % rel(dq, torch.matmul(ds_onlydelta, kf)))
    if S <= 128 and H == 1:
        print("  dq vs dsT@K     %.4f"
              % rel(dq, torch.matmul(ds.transpose(-1, -2), kf)))
        print("  dq vs ds@K^T    %.4f"
              % rel(dq, torch.matmul(ds, kf.transpose(-1, -2))))
    # per-row-block error map (16-row granularity)
    e = (dq.float() - dq_ref).view(B, H, S // 16, 16, 128)
    r = dq_ref.view(B, H, S // 16, 16, 128)
    blk = (e.norm(dim=(-1, -2)) / (r.norm(dim=(-1, -2)) + 1e-8))[0, 0]
    print("  dq err by 16-row block:",
          " ".join("%.2f" % x for x in blk.tolist()))
    ek = (dk.float() - dk_ref).view(B, Hkv, S // 16, 16, 128)
    rk = dk_ref.view(B, Hkv, S // 16, 16, 128)
    blkk = (ek.norm(dim=(-1, -2)) / (rk.norm(dim=(-1, -2)) + 1e-8))[0, 0]
    print("  dk err by 16-row block:",
          " ".join("%.2f" % x for x in blkk.tolist()))


if __name__ == "__main__":
    run_case(1, 1, 1, 64)
    run_case(1, 1, 1, 128)
    run_case(1, 1, 1, 256)
    run_case(1, 4, 2, 256)
