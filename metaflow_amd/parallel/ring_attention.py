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
``j > r`` is skipped entirely.  Plain ``ring_attention`` is therefore
triangular in load; ``zigzag_ring_attention`` below balances it exactly
(every rank does 2W chunk-pair units) by pairing chunk r with its
mirror chunk 2W-1-r on each rank.

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


# ---------------------------------------------------------------- zigzag
def zigzag_shard(x, rank, world, dim=2):
    """Slice a full-sequence tensor into rank r's zigzag shard: the
    sequence is cut into 2W chunks c0..c_{2W-1} and rank r holds
    [c_r ; c_{2W-1-r}].  Pairing an early chunk with its mirror-image
    late chunk gives every rank 2W attention units instead of the plain
    ring's triangular r+1 .. 2W spread."""
    S = x.size(dim)
    c = S // (2 * world)
    early = x.narrow(dim, rank * c, c)
    late = x.narrow(dim, (2 * world - 1 - rank) * c, c)
    return torch.cat([early, late], dim=dim).contiguous()


def zigzag_unshard_grad(parts, rank, world, full_shape, dim=2):
    """Scatter a shard-shaped tensor back into a zero full-sequence
    tensor (used by tests to compare grads)."""
    out = torch.zeros(full_shape, dtype=parts.dtype, device=parts.device)
    S = full_shape[dim]
    c = S // (2 * world)
    out.narrow(dim, rank * c, c).copy_(parts.narrow(dim, 0, c))
    out.narrow(dim, (2 * world - 1 - rank) * c, c).copy_(
        parts.narrow(dim, c, c))
    return out


def _zz_pairs(rank, j, world):
    """Kernel calls for one ring step: rank's q chunks (r, 2W-1-r) vs the
    held kv chunks (j, 2W-1-j). Returns [(q_half, kv_half, causal)]
    with halves 0=early, 1=late. Chunk ki is visible to qi iff ki <= qi;
    every rank ends up with exactly 2W chunk-pair units of work."""
    out = []
    if j == rank:
        out.append((0, 0, True))    # early vs own early: diagonal
        out.append((1, 0, False))   # late sees all of early
        out.append((1, 1, True))    # late vs own late: diagonal
    elif j < rank:
        out.append((0, 0, False))   # q_r sees earlier c_j
        out.append((1, 0, False))   # q_late sees c_j
    else:  # j > rank
        out.append((1, 0, False))   # q_late sees c_j (j<W <= 2W-1-r)
        out.append((1, 1, False))   # q_late sees c_{2W-1-j} (j>r)
    return out


class _ZigzagRingAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale, group):
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        c = q.size(2) // 2
        qh = [q.narrow(2, 0, c), q.narrow(2, c, c)]

        k_cur, v_cur = k, v
        o_acc = [None, None]
        lse_acc = [None, None]
        for step in range(world):
            src = (rank - step) % world
            reqs = []
            if step < world - 1:
                k_nxt = torch.empty_like(k)
                v_nxt = torch.empty_like(v)
                reqs = _ring_post(group, [k_cur, v_cur], [k_nxt, v_nxt])
            for qi, ki, causal in _zz_pairs(rank, src, world):
                o_i, lse_i = K.attn_fwd_raw(
                    qh[qi].contiguous(),
                    k_cur.narrow(2, ki * c, c).contiguous(),
                    v_cur.narrow(2, ki * c, c).contiguous(),
                    scale, causal=causal)
                if o_acc[qi] is None:
                    o_acc[qi] = o_i.float()
                    lse_acc[qi] = lse_i
                else:
                    lse_acc[qi] = _merge(o_acc[qi], lse_acc[qi], o_i,
                                         lse_i)
            if step < world - 1:
                for r_ in reqs:
                    r_.wait()
                k_cur, v_cur = k_nxt, v_nxt
        o = torch.cat([a.to(q.dtype) for a in o_acc], dim=2)
        lse = torch.cat(lse_acc, dim=2)
        ctx.save_for_backward(q, k, v, o, lse)
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
        c = q.size(2) // 2
        qh = [q.narrow(2, 0, c), q.narrow(2, c, c)]
        doh = [dout.narrow(2, 0, c), dout.narrow(2, c, c)]
        oh = [o.narrow(2, 0, c), o.narrow(2, c, c)]
        lseh = [lse.narrow(2, 0, c), lse.narrow(2, c, c)]

        k_cur, v_cur = k, v
        dq_acc = torch.zeros_like(q, dtype=torch.float32)
        dk_ring = torch.zeros_like(k, dtype=torch.float32)
        dv_ring = torch.zeros_like(v, dtype=torch.float32)
        for step in range(world):
            src = (rank - step) % world
            for qi, ki, causal in _zz_pairs(rank, src, world):
                dq_i, dk_i, dv_i = K.attn_bwd_raw(
                    qh[qi].contiguous(),
                    k_cur.narrow(2, ki * c, c).contiguous(),
                    v_cur.narrow(2, ki * c, c).contiguous(),
                    oh[qi].contiguous(), doh[qi].contiguous(),
                    lseh[qi].contiguous(), scale, causal=causal)
                dq_acc.narrow(2, qi * c, c).add_(dq_i.float())
                dk_ring.narrow(2, ki * c, c).add_(dk_i.float())
                dv_ring.narrow(2, ki * c, c).add_(dv_i.float())
            if step < world - 1:
                k_nxt = torch.empty_like(k)
                v_nxt = torch.empty_like(v)
                dk_nxt = torch.empty_like(dk_ring)
                dv_nxt = torch.empty_like(dv_ring)
                _ring_exchange(group, [k_cur, v_cur, dk_ring, dv_ring],
                               [k_nxt, v_nxt, dk_nxt, dv_nxt])
                k_cur, v_cur = k_nxt, v_nxt
                dk_ring, dv_ring = dk_nxt, dv_nxt
        if world > 1:
            dk_home = torch.empty_like(dk_ring)
            dv_home = torch.empty_like(dv_ring)
            _ring_exchange(group, [dk_ring, dv_ring], [dk_home, dv_home])
            dk_ring, dv_ring = dk_home, dv_home
        return (dq_acc.to(q.dtype), dk_ring.to(k.dtype),
                dv_ring.to(v.dtype), None, None)


def zigzag_ring_attention(q, k, v, scale=None, group=None):
    """Load-balanced context-parallel causal GQA flash attention.

    Inputs are zigzag shards (``zigzag_shard``): rank r passes
    ``[c_r ; c_{2W-1-r}]`` of q [B,H,2c,128] and k/v [B,Hkv,2c,128];
    returns the same-shaped output shard, differentiable. Unlike plain
    ``ring_attention`` every rank does identical work (2W chunk-pair
    units), so the ring is not bottlenecked on the last rank.
    """
    if scale is None:
        scale = 1.0 / (q.size(-1) ** 0.5)
    if group is None:
        group = dist.group.WORLD
    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return K.attention(q, k, v, scale)
    return _ZigzagRingAttention.apply(q, k, v, scale, group)
