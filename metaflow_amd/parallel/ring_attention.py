"""Ring attention: context parallelism over the xGMI ring.

Long-context capability beyond the reference: the sequence is
sharded across W ranks (rank r owns token rows ``[r*Sc, (r+1)*Sc)`` of
q/k/v) and KV chunks rotate around the ring while each rank's queries stay
resident.  On MI355X the natural topology IS a ring — xGMI links are
point-to-point (7 × ≈153 GB/s per GPU), so neighbour-to-neighbour KV
rotation overlaps a chunk transfer with a chunk of flash-attention compute
and never congests a switch.

Math: each visible KV chunk produces a partial ``(o_i, lse_i)`` from the
HIP flash kernel (``ops.kernels.attn_fwd_raw``); partials merge by online
softmax::

    lse = log(exp(lse) + exp(lse_i))
    o   = o * exp(lse_old - lse) + o_i * exp(lse_i - lse)

Causality across chunks: a chunk from source rank ``j < r`` is fully
visible (non-causal kernel path), ``j == r`` is the causal diagonal,
``j > r`` is skipped entirely.  (Rank load is therefore triangular; the
zigzag token interleave that balances it is a planned refinement — see
NOTES_ROUND2.md.)

Backward recomputes the ring (flash-style): with the GLOBAL ``(o, lse)``
saved from forward, each chunk's ``attn_bwd_raw`` contribution is exact —
``p = exp(s*scale - lse_global)`` — so dq accumulates locally while
(dk, dv) partials ride the ring W-1 hops until they reach the chunk's
owner.

Works on gloo (CPU tests, isend/irecv) and RCCL alike.
"""

import torch
import torch.distributed as dist

from ..ops import kernels as K


def _ring_post(group, send_tensors, recv_tensors):
    """Post one ring hop (send to next rank, receive from previous);
    returns the requests so the transfer overlaps the caller's compute."""
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    nxt = dist.get_global_rank(group, (rank + 1) % world)
    prv = dist.get_global_rank(group, (rank - 1) % world)
    reqs = []
    # stagger send/recv by parity so neither side blocks on rendezvous
    if rank % 2 == 0:
        reqs += [dist.isend(t, nxt, group=group) for t in send_tensors]
        reqs += [dist.irecv(t, prv, group=group) for t in recv_tensors]
    else:
        reqs += [dist.irecv(t, prv, group=group) for t in recv_tensors]
        reqs += [dist.isend(t, nxt, group=group) for t in send_tensors]
    return reqs


def _ring_exchange(group, send_tensors, recv_tensors):
    for r in _ring_post(group, send_tensors, recv_tensors):
        r.wait()


def _merge(o_acc, lse_acc, o_i, lse_i):
    """Online-softmax merge of a partial (o_i, lse_i) into fp32 o_acc."""
    lse_new = torch.logaddexp(lse_acc, lse_i)
    o_acc.mul_(torch.exp(lse_acc - lse_new).unsqueeze(-1))
    o_acc.add_(o_i.float() * torch.exp(lse_i - lse_new).unsqueeze(-1))
    return lse_new


class _RingAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale, group):
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()

        k_cur, v_cur = k, v
        o_acc = None
        lse_acc = None
        for step in range(world):
            src = (rank - step) % world  # owner of the chunk we hold
            # post the rotation first so the transfer overlaps compute
            reqs = []
            if step < world - 1:
                k_nxt = torch.empty_like(k)
                v_nxt = torch.empty_like(v)
                reqs = _ring_post(group, [k_cur, v_cur], [k_nxt, v_nxt])
            if src <= rank:
                o_i, lse_i = K.attn_fwd_raw(q, k_cur, v_cur, scale,
                                            causal=(src == rank))
                if o_acc is None:
                    o_acc = o_i.float()
                    lse_acc = lse_i
                else:
                    lse_acc = _merge(o_acc, lse_acc, o_i, lse_i)
            if step < world - 1:
                for r in reqs:
                    r.wait()
                k_cur, v_cur = k_nxt, v_nxt
        o = o_acc.to(q.dtype)
        ctx.save_for_backward(q, k, v, o, lse_acc)
        ctx.scale = scale
        ctx.group = group
        return o

    @staticmethod
    def backward(ctx, dout):
        q, k, v, o, lse = ctx.saved_tensors
        scale, group = ctx.scale, ctx.group
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)
        dout = dout.contiguous()

        k_cur, v_cur = k, v
        dk_ring = torch.zeros_like(k, dtype=torch.float32)
        dv_ring = torch.zeros_like(v, dtype=torch.float32)
        dq_acc = torch.zeros_like(q, dtype=torch.float32)
        for step in range(world):
            src = (rank - step) % world
            if step < world - 1:
                k_nxt = torch.empty_like(k)
                v_nxt = torch.empty_like(v)
                dk_nxt = torch.empty_like(dk_ring)
                dv_nxt = torch.empty_like(dv_ring)
            if src <= rank:
                dq_i, dk_i, dv_i = K.attn_bwd_raw(
                    q, k_cur, v_cur, o, dout, lse, scale,
                    causal=(src == rank))
                dq_acc += dq_i.float()
                dk_ring += dk_i.float()
                dv_ring += dv_i.float()
            if step < world - 1:
                # kv and its accumulated grads ride the ring together
                _ring_exchange(group, [k_cur, v_cur, dk_ring, dv_ring],
                               [k_nxt, v_nxt, dk_nxt, dv_nxt])
                k_cur, v_cur = k_nxt, v_nxt
                dk_ring, dv_ring = dk_nxt, dv_nxt
        # after W-1 hops the buffer we hold accumulates grads for the
        # chunk we also hold — which is our own chunk again only if the
        # last hop returned it; rotate once more to bring grads home
        if world > 1:
            dk_home = torch.empty_like(dk_ring)
            dv_home = torch.empty_like(dv_ring)
            _ring_exchange(group, [dk_ring, dv_ring], [dk_home, dv_home])
            dk_ring, dv_ring = dk_home, dv_home
        return (dq_acc.to(q.dtype), dk_ring.to(k.dtype),
                dv_ring.to(v.dtype), None, None)


def ring_attention(q, k, v, scale=None, group=None):
    """Context-parallel causal GQA flash attention.

    Each rank passes its LOCAL sequence shard ``q [B,H,Sc,128]``,
    ``k/v [B,Hkv,Sc,128]`` (token rows ``rank*Sc .. (rank+1)*Sc``); returns
    the local shard of the output, differentiable w.r.t. q/k/v.
    """
    if scale is None:
        scale = 1.0 / (q.size(-1) ** 0.5)
    if group is None:
        group = dist.group.WORLD
    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return K.attention(q, k, v, scale)
    return _RingAttention.apply(q, k, v, scale, group)
