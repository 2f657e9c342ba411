"""Sharper probes: isolate P^T through dv with crafted inputs."""

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


def probe(S=64):
    dev = "cuda:0"
    torch.manual_seed(0)
    scale = 1.0 / math.sqrt(128)
    ext = K.hip_ext()

    q = torch.randn(1, 1, S, 128, dtype=torch.bfloat16, device=dev)
    k = torch.randn(1, 1, S, 128, dtype=torch.bfloat16, device=dev)
    v = torch.zeros(1, 1, S, 128, dtype=torch.bfloat16, device=dev)
    o, lse = ext.attn_fwd(q, k, v, scale)   # o = 0
    dout = torch.randn(1, 1, S, 128, dtype=torch.bfloat16, device=dev)
    dq, dk, dv = ext.attn_bwd(q, k, v, o, dout, lse, scale)
    print("v=0 probe: dq norm %.4f dk norm %.4f (both should be 0)"
          % (dq.float().norm(), dk.float().norm()))

    s = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    mask = torch.ones(S, S, dtype=torch.bool, device=dev).tril()
    p = torch.exp(s - lse.unsqueeze(-1)).masked_fill(~mask, 0.0)
    df = dout.float()
    print("  dv vs P^T dO    %.4f" % rel(dv, torch.matmul(
        p.transpose(-1, -2), df)))
    print("  dv vs P dO      %.4f" % rel(dv, torch.matmul(p, df)))
    pu = torch.exp(s - lse.unsqueeze(-1))  # unmasked
    print("  dv vs Pu^T dO   %.4f" % rel(dv, torch.matmul(
        pu.transpose(-1, -2), df)))
    pm_wrong = torch.exp(s - lse.unsqueeze(-2))  # lse indexed by col
    pm_wrong = pm_wrong.masked_fill(~mask, 0.0)
    print("  dv vs lse-col   %.4f" % rel(dv, torch.matmul(
        pm_wrong.transpose(-1, -2), df)))
    # block map of dv error
    dv_ref = torch.matmul(p.transpose(-1, -2), df)
    e = (dv.float() - dv_ref).view(S // 16, 16, 128)
    r = dv_ref.view(S // 16, 16, 128)
    print("  dv err by kv 16-block:", " ".join(
        "%.2f" % x for x in
        (e.norm(dim=(1, 2)) / (r.norm(dim=(1, 2)) + 1e-8)).tolist()))
    # column blocks (d dimension)
    e2 = (dv.float() - dv_ref).view(S, 8, 16)
    r2 = dv_ref.view(S, 8, 16)
    print("  dv err by d 16-block:", " ".join(
        "%.2f" % x for x in
        (e2.norm(dim=(0, 2)) / (r2.norm(dim=(0, 2)) + 1e-8)).tolist()))

    # q=k=0 probe: S=0 -> P = causal-uniform
    q0 = torch.zeros_like(q)
    k0 = torch.zeros_like(k)
    vr = torch.randn_like(v)
    o0, lse0 = ext.attn_fwd(q0, k0, vr, scale)
    dq0, dk0, dv0 = ext.attn_bwd(q0, k0, vr, o0, dout, lse0, scale)
    pref = torch.ones(S, S, device=dev).tril()
    pref = pref / pref.sum(-1, keepdim=True)
    dv0_ref = torch.matmul(pref.transpose(-1, -2), df[0, 0])
    print("q=k=0 probe: dv vs uniform-P^T dO  %.4f" % rel(dv0, dv0_ref))


if __name__ == "__main__":
    probe(64)
    probe(128)
