"""Llama-3 architecture on the metaflow_amd HIP kernel library.

BASELINE config 3: the @parallel Llama-3-8B bf16 train step. All hot ops
are hand-written gfx950 HIP (ops.attention / rmsnorm / rope / swiglu /
cross_entropy); plain GEMMs go through torch.matmul -> hipBLASLt. Weights
random-init (no network for checkpoints), bf16.
"""

import math
from dataclasses import dataclass

import torch
import torch.nn as nn

from ..ops import kernels as K


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    max_seq_len: int = 8192
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    tie_embeddings: bool = False
    # fp8=True runs the decoder projections' FORWARD GEMMs in OCP E4M3
    # (delayed scaling, bf16 backward — ops/fp8.py); lm_head stays bf16.
    # fp8_bwd=True additionally runs dgrad/wgrad in E4M3 (delayed-scaled
    # dy) — the full-fp8 rung; watch loss quality
    fp8: bool = False
    fp8_bwd: bool = False
    # recompute=True checkpoints each decoder layer's activations
    # (recomputed in backward): activation memory drops from O(layers)
    # to O(1) layers — the knob that fits 70B/long-seq training
    recompute: bool = False

    @classmethod
    def llama3_8b(cls):
        return cls()

    @classmethod
    def llama3_70b(cls):
        return cls(hidden_size=8192, intermediate_size=28672, num_layers=80,
                   num_heads=64, num_kv_heads=8)

    @classmethod
    def tiny(cls, vocab=1024, seq=256):
        """CPU-testable config."""
        return cls(vocab_size=vocab, hidden_size=256, intermediate_size=688,
                   num_layers=2, num_heads=2, num_kv_heads=1, head_dim=128,
                   max_seq_len=seq)


class Linear(nn.Module):
    """Bias-free bf16 linear -> hipBLASLt GEMM (offline-tuned solution
    indices where the shape is in ops/gemm_table.json; heuristic
    otherwise)."""

    def __init__(self, din, dout, dtype=torch.bfloat16):
        super().__init__()
        self.weight = nn.Parameter(
            torch.empty(dout, din, dtype=dtype))

    def forward(self, x):
        from ..ops.gemm import tuned_linear

        return tuned_linear(x, self.weight)


class RMSNorm(nn.Module):
    def __init__(self, dim, eps):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim, dtype=torch.bfloat16))
        self.eps = eps

    def forward(self, x):
        return K.rmsnorm(x, self.weight, self.eps)


class DecoderLayer(nn.Module):
    """Fused projections: one QKV GEMM (the narrow separate K/V GEMMs are
    inefficient on MFMA) and one gate|up GEMM feeding the fused-layout
    SwiGLU kernel (no slice copies; its backward emits d_gateup directly).
    RoPE runs on the [B,H,S,D] layout the attention kernel consumes, so
    the transpose copy after the QKV split is the only data movement."""

    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        h, hd = cfg.hidden_size, cfg.head_dim
        if cfg.fp8:
            import functools

            from ..ops.fp8 import Fp8Linear

            Lin = functools.partial(Fp8Linear, fp8_bwd=cfg.fp8_bwd)
        else:
            Lin = Linear
        self.qkv_proj = Lin(
            h, (cfg.num_heads + 2 * cfg.num_kv_heads) * hd)
        self.input_norm = RMSNorm(h, cfg.rms_eps)
        self.o_proj = Lin(cfg.num_heads * hd, h)
        self.post_norm = RMSNorm(h, cfg.rms_eps)
        self.gate_up_proj = Lin(h, 2 * cfg.intermediate_size)
        self.down_proj = Lin(cfg.intermediate_size, h)

    def forward(self, res, pending, cos_t, sin_t, cp_group=None,
                cache=None, layer_idx=0):
        """Carries (residual_stream, pending_branch): every residual add
        fuses with the next RMSNorm (K.add_rmsnorm). `pending` is the
        previous layer's un-added MLP output (None for layer 0). With
        cp_group set, S is the LOCAL sequence shard and attention runs
        the xGMI ring (parallel/ring_attention.py)."""
        cfg = self.cfg
        B, S, _ = res.shape
        hd = cfg.head_dim
        nq, nkv = cfg.num_heads, cfg.num_kv_heads
        res, y = K.add_rmsnorm(res, pending, self.input_norm.weight,
                               self.input_norm.eps)
        qkv = self.qkv_proj(y)
        # one fused kernel: QKV split + bhsd transpose + RoPE
        q, k, v = K.rope_qkv(qkv, cos_t, sin_t, nq, nkv)
        if cache is not None:
            kc, vc, pos = cache.k[layer_idx], cache.v[layer_idx], \
                cache.pos
            kc[:, :, pos:pos + S] = k
            vc[:, :, pos:pos + S] = v
            if pos == 0:
                o = K.attention(q, k, v)      # prefill: flash (padded
                # to the 256 tile internally for unaligned prompts)
            elif S == 1:
                # one-token decode: split-K flash-decode straight over
                # the cache allocation (decode.hip)
                o = K.attn_decode(q, kc, vc, pos + 1,
                                  1.0 / math.sqrt(hd))
            else:
                o = _attn_with_cache(q, kc[:, :, :pos + S],
                                     vc[:, :, :pos + S],
                                     1.0 / math.sqrt(hd), pos)
        elif cp_group is not None:
            from ..parallel.ring_attention import ring_attention

            o = ring_attention(q, k, v, group=cp_group)
        else:
            o = K.attention(q, k, v)
        o = o.transpose(1, 2).reshape(B, S, nq * hd)

        res, y = K.add_rmsnorm(res, self.o_proj(o), self.post_norm.weight,
                               self.post_norm.eps)
        pending = self.down_proj(K.swiglu_fused(self.gate_up_proj(y)))
        return res, pending

    def decode_step(self, res, pending, c_rows, s_rows, cache, layer_idx,
                    pos_t, lengths, max_len=None):
        """Continuous-batching decode: one new token per slot, each at
        its OWN cache position (pos_t [B]); the attention is the varlen
        flash-decode kernel over the full cache allocation."""
        cfg = self.cfg
        B = res.size(0)
        hd = cfg.head_dim
        nq, nkv = cfg.num_heads, cfg.num_kv_heads
        res, y = K.add_rmsnorm(res, pending, self.input_norm.weight,
                               self.input_norm.eps)
        qkv = self.qkv_proj(y)
        q, k, v = qkv.split([nq * hd, nkv * hd, nkv * hd], dim=-1)
        q = q.view(B, 1, nq, hd).transpose(1, 2)     # [B, nq, 1, hd]
        k = k.view(B, 1, nkv, hd).transpose(1, 2)
        v = v.view(B, 1, nkv, hd).transpose(1, 2)
        q = _rope_rows(q, c_rows, s_rows)
        k = _rope_rows(k, c_rows, s_rows)
        kc, vc = cache.k[layer_idx], cache.v[layer_idx]
        bidx = torch.arange(B, device=res.device)
        kc[bidx, :, pos_t] = k[:, :, 0]
        vc[bidx, :, pos_t] = v[:, :, 0]
        o = K.attn_decode_varlen(q.contiguous(), kc, vc, lengths,
                                 1.0 / math.sqrt(hd), max_len=max_len)
        o = o.transpose(1, 2).reshape(B, 1, nq * hd)
        res, y = K.add_rmsnorm(res, self.o_proj(o), self.post_norm.weight,
                               self.post_norm.eps)
        pending = self.down_proj(K.swiglu_fused(self.gate_up_proj(y)))
        return res, pending


def _rope_rows(x, c, s):
    """Half-rotation RoPE with a per-BATCH position row: x [B,Hh,1,D],
    c/s [B, D/2] (each slot's own position — the batched-decode case the
    positional kernel's single-offset layout can't express)."""
    B, _hh, _one, D = x.shape
    half = D // 2
    c = c.view(B, 1, 1, half).float()
    s = s.view(B, 1, 1, half).float()
    xf = x.float()
    x1, x2 = xf[..., :half], xf[..., half:]
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], -1).to(x.dtype)


class KVCache(object):
    """Preallocated per-layer KV cache for autoregressive decode.

    k/v: one [B, Hkv, max_len, 128] bf16 tensor per layer, filled
    in-place as positions arrive; ``pos`` is the number of cached
    positions (advanced once per model forward, not per layer).
    """

    def __init__(self, cfg, batch, max_len, device, dtype=torch.bfloat16):
        self.k = [torch.zeros(batch, cfg.num_kv_heads, max_len,
                              cfg.head_dim, device=device, dtype=dtype)
                  for _ in range(cfg.num_layers)]
        self.v = [torch.zeros_like(k) for k in self.k]
        self.pos = 0
        self.max_len = max_len


def _attn_with_cache(q, k_all, v_all, scale, pos):
    """Attention of the s new queries (global positions pos..pos+s)
    against all T cached kv positions: full visibility of the prefix,
    causal within the new block. Plain torch ops (fp32 softmax) — decode
    blocks are tiny, the flash kernels handle the big prefill."""
    B, H, s, D = q.shape
    Hkv = k_all.size(1)
    T = k_all.size(2)
    G = H // Hkv
    kf = k_all.float().repeat_interleave(G, dim=1)
    vf = v_all.float().repeat_interleave(G, dim=1)
    att = torch.matmul(q.float(), kf.transpose(-1, -2)) * scale
    kv_pos = torch.arange(T, device=q.device)
    q_pos = pos + torch.arange(s, device=q.device)
    mask = kv_pos[None, :] <= q_pos[:, None]          # [s, T]
    att = att.masked_fill(~mask, float("-inf"))
    p = torch.softmax(att, dim=-1)
    return torch.matmul(p, vf).to(q.dtype)


class LlamaForCausalLM(nn.Module):
    """Set ``cp_group`` (a torch.distributed group) for context-parallel
    training: every rank in the group holds the SAME parameters but a
    different contiguous sequence shard of each batch (rank r gets token
    rows r*Sc..(r+1)*Sc); attention runs over the xGMI ring. Parameters
    being replicated, the per-rank local-mean losses backprop to grads
    whose cp-group AVERAGE is the true global-mean-loss gradient — i.e.
    cp ranks join the flat-buffer DDP all-reduce exactly like extra data-
    parallel ranks (parallel/ddp.py needs no changes)."""

    #: decode_step is free of data-dependent shapes -> the serving
    #: engine may capture it in a hipGraph and replay per token
    graph_safe_decode = True

    def __init__(self, cfg: LlamaConfig, cp_group=None):
        super().__init__()
        self.cfg = cfg
        self.cp_group = cp_group
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size,
                                  dtype=torch.bfloat16)
        self.layers = nn.ModuleList(
            DecoderLayer(cfg) for _ in range(cfg.num_layers))
        self.final_norm = RMSNorm(cfg.hidden_size, cfg.rms_eps)
        self.lm_head = Linear(cfg.hidden_size, cfg.vocab_size)
        cos_t, sin_t = K.rope_tables(cfg.max_seq_len, cfg.head_dim,
                                     cfg.rope_theta)
        self.register_buffer("cos_t", cos_t, persistent=False)
        self.register_buffer("sin_t", sin_t, persistent=False)
        self.reset_parameters()

    def reset_parameters(self):
        std = 0.02
        for name, p in self.named_parameters():
            if p.dim() >= 2:
                nn.init.normal_(p, mean=0.0, std=std)
            elif "norm" in name:
                nn.init.ones_(p)
        # scaled init for output projections (GPT-2 style)
        scale = 1.0 / math.sqrt(2 * self.cfg.num_layers)
        for layer in self.layers:
            layer.o_proj.weight.data.mul_(scale)
            layer.down_proj.weight.data.mul_(scale)

    def forward(self, tokens, targets=None, cache=None):
        """tokens [B, S] int64 (the LOCAL shard when cp_group is set);
        returns mean loss over the local tokens if targets given, else
        logits. With ``cache`` (a KVCache), S may be a suffix of an
        ongoing sequence starting at position cache.pos."""
        S = tokens.size(1)
        off = cache.pos if cache is not None else 0
        if self.cp_group is not None:
            import torch.distributed as dist

            off = dist.get_rank(self.cp_group) * S
        cos_t = self.cos_t[off:off + S].contiguous() if off else self.cos_t
        sin_t = self.sin_t[off:off + S].contiguous() if off else self.sin_t
        res = self.embed(tokens)
        pending = None
        recompute = (self.cfg.recompute and torch.is_grad_enabled()
                     and cache is None)
        for li, layer in enumerate(self.layers):
            if recompute:
                from torch.utils.checkpoint import checkpoint

                res, pending = checkpoint(
                    layer, res, pending, cos_t, sin_t, self.cp_group,
                    cache, li, use_reentrant=False)
            else:
                res, pending = layer(res, pending, cos_t, sin_t,
                                     self.cp_group, cache, li)
        if cache is not None:
            cache.pos += S
        _, x = K.add_rmsnorm(res, pending, self.final_norm.weight,
                             self.final_norm.eps)
        if targets is None:
            return self.lm_head(x)
        logits = self.lm_head(x)
        B, S, V = logits.shape
        loss = K.cross_entropy(logits.reshape(B * S, V),
                               targets.reshape(B * S))
        return loss.mean()

    @torch.no_grad()
    def decode_step(self, tokens, cache, positions, max_len=None):
        """One batched decode step for continuous batching: tokens
        [B, 1], positions[b] = this token's cache position for slot b
        (slots may sit at DIFFERENT positions; inactive slots pass 0 and
        their output row is garbage the batcher discards). Returns
        logits [B, 1, V].

        With ``max_len`` (the cache capacity) the whole step is free of
        host reads and is hipGraph-capturable: the serving engine
        captures it once and replays per token (decode is launch-bound;
        see serving.ContinuousBatcher(graph=True))."""
        pos_t = torch.as_tensor(positions, device=tokens.device,
                                dtype=torch.long)
        lengths = (pos_t + 1).to(torch.int32)
        c_rows = self.cos_t[pos_t]
        s_rows = self.sin_t[pos_t]
        res = self.embed(tokens)
        pending = None
        for li, layer in enumerate(self.layers):
            res, pending = layer.decode_step(res, pending, c_rows,
                                             s_rows, cache, li, pos_t,
                                             lengths, max_len=max_len)
        _, x = K.add_rmsnorm(res, pending, self.final_norm.weight,
                             self.final_norm.eps)
        return self.lm_head(x)

    @torch.no_grad()
    def generate(self, tokens, max_new_tokens, temperature=0.0,
                 top_k=None):
        """Autoregressive decode with a KV cache: flash-kernel prefill
        (when the prompt length is 64-aligned), cached attention for the
        one-token decode steps. temperature=0 is greedy."""
        B, S0 = tokens.shape
        cache = KVCache(self.cfg, B, S0 + max_new_tokens,
                        tokens.device,
                        dtype=self.embed.weight.dtype)
        out = tokens
        logits = self.forward(tokens, cache=cache)
        for _ in range(max_new_tokens):
            last = logits[:, -1].float()
            if temperature and temperature > 0:
                last = last / temperature
                if top_k:
                    kth = torch.topk(last, top_k, dim=-1).values[:, -1:]
                    last = last.masked_fill(last < kth, float("-inf"))
                probs = torch.softmax(last, dim=-1)
                nxt = torch.multinomial(probs, 1)
            else:
                nxt = last.argmax(dim=-1, keepdim=True)
            out = torch.cat([out, nxt], dim=1)
            logits = self.forward(nxt, cache=cache)
        return out

    def num_params(self):
        return sum(p.numel() for p in self.parameters())
