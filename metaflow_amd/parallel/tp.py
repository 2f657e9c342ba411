"""Tensor parallelism over xGMI: Megatron-style sharded projections.

The BASELINE north-star names DP/TP as the in-node parallelisms; this
module provides the TP half. On MI355X, TP lives INSIDE a node: every
transformer layer costs two all-reduces (attention output + MLP output)
of [B,S,h] activations over the tp group — xGMI's 7×153 GB/s
point-to-point links make tp=2..8 viable when a model (70B+) outgrows
even 288 GB of HBM per GPU or when activation memory dominates.

Building blocks (each a pair of autograd-transposed collectives):

* ``copy_to_tp``      — forward identity, backward all-reduce (the input
  of a column-parallel GEMM is consumed by every rank, so its grads sum)
* ``reduce_from_tp``  — forward all-reduce, backward identity (the
  output of a row-parallel GEMM is a partial sum)
* ``gather_from_tp``  — forward all-gather along the last dim, backward
  take-my-slice (column-sharded lm_head logits)

`ColumnParallelLinear` shards output features, `RowParallelLinear`
shards input features; chained column→row gives one all-reduce per MLP
and per attention block. `shard_*` helpers cut a FULL weight into rank
shards so a TP model can be built to match a reference model exactly
(used by the gloo equivalence tests)."""

import torch
import torch.distributed as dist
import torch.nn as nn


class _CopyToTP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad):
        g = grad.contiguous()
        dist.all_reduce(g, group=ctx.group)
        return g, None


class _ReduceFromTP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        x = x.contiguous()
        dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):
        return grad, None


class _GatherFromTP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        world = dist.get_world_size(group)
        rank = dist.get_rank(group)
        ctx.group = group
        ctx.rank = rank
        ctx.world = world
        ctx.dim_local = x.size(-1)
        parts = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(parts, x.contiguous(), group=group)
        parts[rank] = x  # keep the autograd-local copy in place
        return torch.cat(parts, dim=-1)

    @staticmethod
    def backward(ctx, grad):
        lo = ctx.rank * ctx.dim_local
        return grad[..., lo:lo + ctx.dim_local].contiguous(), None


def copy_to_tp(x, group):
    return _CopyToTP.apply(x, group)


def reduce_from_tp(x, group):
    return _ReduceFromTP.apply(x, group)


def gather_from_tp(x, group):
    return _GatherFromTP.apply(x, group)


class ColumnParallelLinear(nn.Module):
    """y_shard = x @ W_shardᵀ: output features sharded, input replicated
    (grads of the input all-reduce in backward via copy_to_tp)."""

    def __init__(self, din, dout, group, dtype=torch.bfloat16):
        super().__init__()
        self.group = group
        world = dist.get_world_size(group)
        assert dout % world == 0, (dout, world)
        self.weight = nn.Parameter(
            torch.empty(dout // world, din, dtype=dtype))

    def forward(self, x):
        x = copy_to_tp(x, self.group)
        return torch.nn.functional.linear(x, self.weight)


class RowParallelLinear(nn.Module):
    """y = all_reduce(x_shard @ W_shardᵀ): input features sharded,
    output replicated."""

    def __init__(self, din, dout, group, dtype=torch.bfloat16):
        super().__init__()
        self.group = group
        world = dist.get_world_size(group)
        assert din % world == 0, (din, world)
        self.weight = nn.Parameter(
            torch.empty(dout, din // world, dtype=dtype))

    def forward(self, x_shard):
        partial = torch.nn.functional.linear(x_shard, self.weight)
        return reduce_from_tp(partial, self.group)


# ------------------------------------------------------ shard helpers
def shard_rows(full, rank, world):
    """Column-parallel shard: rows [r*dout/w, (r+1)*dout/w) of W."""
    n = full.size(0) // world
    return full[rank * n:(rank + 1) * n].clone()


def shard_cols(full, rank, world):
    """Row-parallel shard: input-feature columns of W."""
    n = full.size(1) // world
    return full[:, rank * n:(rank + 1) * n].clone()


def shard_qkv_rows(full, rank, world, nq, nkv, hd):
    """Shard a fused QKV weight [(nq+2*nkv)*hd, din] by HEADS: rank r
    takes q-heads [r*nq/w..), k-heads and v-heads likewise, keeping the
    fused q|k|v row order within the shard."""
    qn, kn = nq * hd, nkv * hd
    q, k, v = full[:qn], full[qn:qn + kn], full[qn + kn:]
    return torch.cat([shard_rows(q, rank, world),
                      shard_rows(k, rank, world),
                      shard_rows(v, rank, world)], dim=0)


def shard_gate_up_rows(full, rank, world):
    """Shard a fused gate|up weight [2*I, din]: each half sharded
    independently so the local layout stays gate_local|up_local (what
    the fused SwiGLU kernel consumes)."""
    half = full.size(0) // 2
    return torch.cat([shard_rows(full[:half], rank, world),
                      shard_rows(full[half:], rank, world)], dim=0)


class _VocabParallelCE(torch.autograd.Function):
    """Cross-entropy over a vocab-sharded logits tensor WITHOUT
    materializing the full logits: the softmax statistics (max,
    sum-exp) and the target logit are each one scalar-per-token
    all-reduce — O(N) communication instead of O(N*V) for an
    all-gather. Returns per-token loss [N]."""

    @staticmethod
    def forward(ctx, logits_shard, targets, group, vocab_start):
        lf = logits_shard.float()
        N, Vl = lf.shape
        m = lf.max(dim=-1).values
        dist.all_reduce(m, op=dist.ReduceOp.MAX, group=group)
        se = torch.exp(lf - m[:, None]).sum(dim=-1)
        dist.all_reduce(se, group=group)
        lse = m + torch.log(se)

        local = targets - vocab_start
        in_range = (local >= 0) & (local < Vl)
        idx = local.clamp(0, Vl - 1)
        tgt_logit = torch.where(
            in_range, lf.gather(1, idx[:, None]).squeeze(1),
            torch.zeros_like(m))
        dist.all_reduce(tgt_logit, group=group)
        loss = lse - tgt_logit
        ctx.save_for_backward(logits_shard, targets, lse)
        ctx.group = group
        ctx.vocab_start = vocab_start
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits_shard, targets, lse = ctx.saved_tensors
        lf = logits_shard.float()
        N, Vl = lf.shape
        p = torch.exp(lf - lse[:, None])
        local = targets - ctx.vocab_start
        in_range = (local >= 0) & (local < Vl)
        idx = local.clamp(0, Vl - 1)
        p.scatter_add_(
            1, idx[:, None],
            torch.where(in_range, -torch.ones_like(lse),
                        torch.zeros_like(lse))[:, None])
        dlogits = (p * dloss.float()[:, None]).to(logits_shard.dtype)
        return dlogits, None, None, None


def vocab_parallel_cross_entropy(logits_shard, targets, group,
                                 vocab_start):
    """Per-token loss [N] from vocab-sharded logits [N, V/w]."""
    return _VocabParallelCE.apply(logits_shard.contiguous(), targets,
                                  group, vocab_start)


# ----------------------------------------------------- sequence parallel
# Megatron-style SP: the activations BETWEEN sharded GEMMs (norms,
# residual adds) are sequence-sharded [B, S/w, h] instead of replicated,
# cutting their memory by w. The tp all-reduces become
# all-gather (before a column GEMM) / reduce-scatter (after a row GEMM)
# pairs of identical total volume. reduce-scatter is emulated as
# all-reduce + take-my-slice so the same code runs gloo (CPU tests) and
# RCCL (which has the real collective — round-2 swap-in).

def _sp_slice(x, group):
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    sc = x.size(1) // world
    return x[:, rank * sc:(rank + 1) * sc].contiguous()


class _AllGatherSP(torch.autograd.Function):
    """[B, S/w, h] -> [B, S, h]; backward reduce-scatters the grads."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        world = dist.get_world_size(group)
        x = x.contiguous()
        parts = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(parts, x, group=group)
        parts[dist.get_rank(group)] = x
        return torch.cat(parts, dim=1)

    @staticmethod
    def backward(ctx, grad):
        g = grad.contiguous()
        dist.all_reduce(g, group=ctx.group)
        return _sp_slice(g, ctx.group), None


class _ReduceScatterSP(torch.autograd.Function):
    """Sum partials over the group and keep this rank's sequence chunk:
    [B, S, h] -> [B, S/w, h]; backward all-gathers."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        g = x.contiguous()
        dist.all_reduce(g, group=group)
        return _sp_slice(g, group)

    @staticmethod
    def backward(ctx, grad):
        world = dist.get_world_size(ctx.group)
        g = grad.contiguous()
        parts = [torch.empty_like(g) for _ in range(world)]
        dist.all_gather(parts, g, group=ctx.group)
        parts[dist.get_rank(ctx.group)] = g
        return torch.cat(parts, dim=1), None


def all_gather_sp(x, group):
    return _AllGatherSP.apply(x, group)


def reduce_scatter_sp(x, group):
    return _ReduceScatterSP.apply(x, group)


def mark_sp_partial(*params):
    """Tag params whose grads are partial under SP (they only saw this
    rank's sequence chunk): norms, embeddings."""
    for p in params:
        p._mfx_sp_partial = True


def sp_sync_grads(model, group):
    """All-reduce the grads of SP-partial params over the tp group —
    call after backward, before the optimizer (the analogue of
    Megatron's sequence-parallel grad sync)."""
    for p in model.parameters():
        if getattr(p, "_mfx_sp_partial", False) and p.grad is not None:
            dist.all_reduce(p.grad, group=group)
