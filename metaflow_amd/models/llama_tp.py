"""Tensor-parallel Llama: Megatron-style head/feature sharding on the
same HIP kernel library as models/llama.py.

Per layer, rank r holds nq/w query heads + nkv/w kv heads (attention is
fully local to the rank's heads — GQA ratio preserved), a gate|up shard
of the MLP, and row shards of o_proj/down_proj; exactly TWO all-reduces
of [B, S, h] per layer (attention output + MLP output) travel the tp
group over xGMI. The lm_head is vocab-sharded and the loss is a
vocab-parallel cross-entropy — three scalar-per-token all-reduces
instead of gathering [B,S,V] logits.

Composes with the other axes: put tp inside the gang's innermost ranks
and pass the dp group to FlatParamModel(group=...) — sharded params
must only all-reduce across ranks holding the SAME shard.

`from_full_model` builds a TP model holding exact shards of a reference
`LlamaForCausalLM` — the gloo equivalence tests
(tests/test_tensor_parallel.py) check loss and every weight gradient
against the unsharded model.
"""

import math

import torch
import torch.nn as nn

from ..ops import kernels as K
from ..parallel.tp import (
    ColumnParallelLinear,
    RowParallelLinear,
    all_gather_sp,
    gather_from_tp,
    mark_sp_partial,
    reduce_scatter_sp,
    shard_cols,
    shard_gate_up_rows,
    shard_qkv_rows,
    vocab_parallel_cross_entropy,
)
from .llama import LlamaConfig, RMSNorm


class TPDecoderLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig, group, sequence_parallel=False):
        super().__init__()
        import torch.distributed as dist

        self.cfg = cfg
        self.group = group
        self.sp = sequence_parallel
        self.world = dist.get_world_size(group)
        h, hd = cfg.hidden_size, cfg.head_dim
        assert cfg.num_heads % self.world == 0
        assert cfg.num_kv_heads % self.world == 0
        assert cfg.intermediate_size % self.world == 0
        self.nq_l = cfg.num_heads // self.world
        self.nkv_l = cfg.num_kv_heads // self.world
        self.qkv_proj = ColumnParallelLinear(
            h, (cfg.num_heads + 2 * cfg.num_kv_heads) * hd, group)
        self.input_norm = RMSNorm(h, cfg.rms_eps)
        self.o_proj = RowParallelLinear(cfg.num_heads * hd, h, group)
        self.post_norm = RMSNorm(h, cfg.rms_eps)
        self.gate_up_proj = ColumnParallelLinear(
            h, 2 * cfg.intermediate_size, group)
        self.down_proj = RowParallelLinear(cfg.intermediate_size, h, group)

    def forward(self, res, pending, cos_t, sin_t):
        """Without SP, res/pending are replicated [B,S,h]; with SP they
        are sequence shards [B,S/w,h] and every column GEMM is preceded
        by an all-gather, every row GEMM followed by a reduce-scatter
        (same comm volume as the plain tp all-reduces, 1/w the
        norm/residual activation memory)."""
        cfg = self.cfg
        hd = cfg.head_dim
        res, y = K.add_rmsnorm(res, pending, self.input_norm.weight,
                               self.input_norm.eps)
        if self.sp:
            y = all_gather_sp(y, self.group)
            qkv = torch.nn.functional.linear(y, self.qkv_proj.weight)
        else:
            qkv = self.qkv_proj(y)      # local heads only
        B, S, _ = qkv.shape
        q, k, v = K.rope_qkv(qkv, cos_t, sin_t, self.nq_l, self.nkv_l)
        o = K.attention(q, k, v)        # local GQA group, no comm
        o = o.transpose(1, 2).reshape(B, S, self.nq_l * hd)
        if self.sp:
            att = reduce_scatter_sp(
                torch.nn.functional.linear(o, self.o_proj.weight),
                self.group)
        else:
            att = self.o_proj(o)
        res, y = K.add_rmsnorm(res, att, self.post_norm.weight,
                               self.post_norm.eps)
        if self.sp:
            y = all_gather_sp(y, self.group)
            gu = torch.nn.functional.linear(y, self.gate_up_proj.weight)
            pending = reduce_scatter_sp(
                torch.nn.functional.linear(
                    K.swiglu_fused(gu), self.down_proj.weight),
                self.group)
        else:
            pending = self.down_proj(K.swiglu_fused(self.gate_up_proj(y)))
        return res, pending


class TPLlamaForCausalLM(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp_group,
                 sequence_parallel=False):
        super().__init__()
        self.cfg = cfg
        self.group = tp_group
        self.sp = sequence_parallel
        import torch.distributed as dist

        self.world = dist.get_world_size(tp_group)
        assert cfg.vocab_size % self.world == 0
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size,
                                  dtype=torch.bfloat16)
        self.layers = nn.ModuleList(
            TPDecoderLayer(cfg, tp_group, sequence_parallel)
            for _ in range(cfg.num_layers))
        self.final_norm = RMSNorm(cfg.hidden_size, cfg.rms_eps)
        self.lm_head = ColumnParallelLinear(cfg.hidden_size,
                                            cfg.vocab_size, tp_group)
        cos_t, sin_t = K.rope_tables(cfg.max_seq_len, cfg.head_dim,
                                     cfg.rope_theta)
        self.register_buffer("cos_t", cos_t, persistent=False)
        self.register_buffer("sin_t", sin_t, persistent=False)
        self.reset_parameters()
        if sequence_parallel:
            # these params only see this rank's sequence chunk: their
            # grads need sp_sync_grads(model, tp_group) after backward
            mark_sp_partial(self.embed.weight, self.final_norm.weight,
                            *[l.input_norm.weight for l in self.layers],
                            *[l.post_norm.weight for l in self.layers])

    def reset_parameters(self):
        std = 0.02
        for name, p in self.named_parameters():
            if p.dim() >= 2:
                nn.init.normal_(p, mean=0.0, std=std)
            elif "norm" in name:
                nn.init.ones_(p)
        scale = 1.0 / math.sqrt(2 * self.cfg.num_layers)
        for layer in self.layers:
            layer.o_proj.weight.data.mul_(scale)
            layer.down_proj.weight.data.mul_(scale)

    @classmethod
    def from_full_model(cls, full, tp_group, sequence_parallel=False):
        """Build the TP model holding exact shards of an unsharded
        LlamaForCausalLM (equivalence tests / converting checkpoints)."""
        import torch.distributed as dist

        cfg = full.cfg
        rank = dist.get_rank(tp_group)
        world = dist.get_world_size(tp_group)
        m = cls(cfg, tp_group, sequence_parallel)
        with torch.no_grad():
            m.embed.weight.copy_(full.embed.weight)
            m.final_norm.weight.copy_(full.final_norm.weight)
            m.lm_head.weight.copy_(
                full.lm_head.weight[rank * (cfg.vocab_size // world):
                                    (rank + 1) * (cfg.vocab_size // world)])
            for tl, fl in zip(m.layers, full.layers):
                tl.input_norm.weight.copy_(fl.input_norm.weight)
                tl.post_norm.weight.copy_(fl.post_norm.weight)
                tl.qkv_proj.weight.copy_(shard_qkv_rows(
                    fl.qkv_proj.weight, rank, world, cfg.num_heads,
                    cfg.num_kv_heads, cfg.head_dim))
                tl.o_proj.weight.copy_(shard_cols(
                    fl.o_proj.weight, rank, world))
                tl.gate_up_proj.weight.copy_(shard_gate_up_rows(
                    fl.gate_up_proj.weight, rank, world))
                tl.down_proj.weight.copy_(shard_cols(
                    fl.down_proj.weight, rank, world))
        return m

    def forward(self, tokens, targets=None):
        import torch.distributed as dist

        S = tokens.size(1)
        cos_t, sin_t = self.cos_t, self.sin_t
        res = self.embed(tokens)
        if self.sp:  # each rank keeps its sequence chunk
            sc = S // self.world
            r = dist.get_rank(self.group)
            res = res[:, r * sc:(r + 1) * sc].contiguous()
        pending = None
        for layer in self.layers:
            res, pending = layer(res, pending, cos_t, sin_t)
        _, x = K.add_rmsnorm(res, pending, self.final_norm.weight,
                             self.final_norm.eps)
        if self.sp:  # back to the full sequence for the lm_head
            # NOTE: all_gather_sp's backward already sums input grads
            # over the group, so bypass copy_to_tp (double reduction)
            x = all_gather_sp(x, self.group)
            logits_shard = torch.nn.functional.linear(
                x, self.lm_head.weight)
        else:
            logits_shard = self.lm_head(x)
        if targets is None:
            return gather_from_tp(logits_shard, self.group)
        import torch.distributed as dist

        B, S, Vl = logits_shard.shape
        v0 = dist.get_rank(self.group) * Vl
        # vocab-parallel CE: O(tokens) comm, the full [B,S,V] logits are
        # never materialized (tp.py: _VocabParallelCE)
        loss = vocab_parallel_cross_entropy(
            logits_shard.reshape(B * S, Vl), targets.reshape(B * S),
            self.group, v0)
        return loss.mean()

    def num_params(self):
        return sum(p.numel() for p in self.parameters())
