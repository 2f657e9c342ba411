"""Pipeline-parallel Llama: layer stages across ranks, GPipe fill-drain.

Completes the DP/TP/PP triad (BASELINE north-star): stage r holds layers
[r*L/w, (r+1)*L/w) — stage 0 adds the embedding, the last stage the
final norm + lm_head + loss. Between stages only the (residual,
pending) activation pair travels (bf16 [B_m, S, h] x2 per micro-batch
hop), point-to-point over one xGMI link — the natural placement is
consecutive GPUs so each boundary is a dedicated link.

`pp_train_step` runs the schedule: forward all M micro-batches through
the pipe (send/recv activations), then backward in reverse order,
shipping boundary gradients back. Micro-batching bounds the pipeline
bubble at (w-1)/(M+w-1); the default 1F1B schedule also bounds live
activations at pipeline-depth micro-batches (fill-drain available as
schedule="gpipe").

Works on gloo (CPU equivalence tests vs the unsharded model —
micro-batch=1 is bit-identical compute) and RCCL alike.
"""

import math

import torch
import torch.distributed as dist
import torch.nn as nn

from ..ops import kernels as K
from .llama import DecoderLayer, LlamaConfig, RMSNorm


class PPLlamaStage(nn.Module):
    def __init__(self, cfg: LlamaConfig, pp_group):
        super().__init__()
        self.cfg = cfg
        self.group = pp_group
        self.world = dist.get_world_size(pp_group)
        self.stage = dist.get_rank(pp_group)
        assert cfg.num_layers % self.world == 0
        per = cfg.num_layers // self.world
        self.layer_lo = self.stage * per
        self.is_first = self.stage == 0
        self.is_last = self.stage == self.world - 1
        if self.is_first:
            self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size,
                                      dtype=torch.bfloat16)
        self.layers = nn.ModuleList(DecoderLayer(cfg) for _ in range(per))
        if self.is_last:
            self.final_norm = RMSNorm(cfg.hidden_size, cfg.rms_eps)
            self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size,
                                     bias=False, dtype=torch.bfloat16)
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
        scale = 1.0 / math.sqrt(2 * self.cfg.num_layers)
        for layer in self.layers:
            layer.o_proj.weight.data.mul_(scale)
            layer.down_proj.weight.data.mul_(scale)

    @classmethod
    def from_full_model(cls, full, pp_group):
        """This rank's stage holding exact copies of the unsharded
        model's layer range (equivalence tests / checkpoint split)."""
        cfg = full.cfg
        m = cls(cfg, pp_group)
        with torch.no_grad():
            if m.is_first:
                m.embed.weight.copy_(full.embed.weight)
            if m.is_last:
                m.final_norm.weight.copy_(full.final_norm.weight)
                m.lm_head.weight.copy_(full.lm_head.weight)
            for i, layer in enumerate(m.layers):
                src = full.layers[m.layer_lo + i]
                layer.load_state_dict(src.state_dict())
        return m

    def forward_stage(self, res, pending):
        """Run this stage's layers on an incoming (res, pending) pair."""
        for layer in self.layers:
            res, pending = layer(res, pending, self.cos_t, self.sin_t)
        return res, pending

    def forward_tokens(self, tokens):
        assert self.is_first
        return self.forward_stage(self.embed(tokens), None)

    def loss_head(self, res, pending, targets):
        assert self.is_last
        _, x = K.add_rmsnorm(res, pending, self.final_norm.weight,
                             self.final_norm.eps)
        logits = self.lm_head(x)
        B, S, V = logits.shape
        loss = K.cross_entropy(logits.reshape(B * S, V),
                               targets.reshape(B * S))
        return loss.mean()

    def num_params(self):
        return sum(p.numel() for p in self.parameters())


def _recv(shape, dtype, src, group, device):
    t = torch.empty(shape, dtype=dtype, device=device)
    dist.recv(t, src, group=group)
    return t


def pp_train_step(stage_model, tokens, targets, microbatches=1,
                  schedule="1f1b"):
    """One pipeline step; every rank returns the mean loss.

    tokens/targets are the FULL batch [B, S] on every rank (only the
    first/last stages read them); grads accumulate into the stage's
    params — run your optimizer afterwards.

    schedule="1f1b" (default) runs the one-forward-one-backward steady
    state: at most (pipeline_depth - stage) micro-batch activations are
    ever live, so M can grow (shrinking the (w-1)/(M+w-1) bubble)
    without growing memory. "gpipe" is the plain fill-drain (all M
    forwards, then all M backwards) — same numerics, simpler trace.
    """
    m = stage_model
    group = m.group
    cfg = m.cfg
    dev = next(m.parameters()).device
    nxt = dist.get_global_rank(group, m.stage + 1) if not m.is_last \
        else None
    prv = dist.get_global_rank(group, m.stage - 1) if not m.is_first \
        else None

    B, S = tokens.shape
    assert B % microbatches == 0
    mb = B // microbatches
    shape = (mb, S, cfg.hidden_size)

    from collections import deque

    saved = deque()     # FIFO of (inputs, outputs) awaiting backward
    losses = []
    sends = []          # in-flight isends (keep refs until waited)
    state = {"fwd": 0, "bwd": 0}

    def _isend(t, dst):
        sends.append(dist.isend(t.contiguous(), dst, group=group))

    def do_fwd():
        i = state["fwd"]
        sl = slice(i * mb, (i + 1) * mb)
        if m.is_first:
            inp = None
            res, pending = m.forward_tokens(tokens[sl])
        else:
            r_in = _recv(shape, torch.bfloat16, prv, group,
                         dev).requires_grad_(True)
            p_in = _recv(shape, torch.bfloat16, prv, group,
                         dev).requires_grad_(True)
            inp = (r_in, p_in)
            res, pending = m.forward_stage(r_in, p_in)
        if m.is_last:
            loss = m.loss_head(res, pending, targets[sl])
            losses.append(loss)
            saved.append((inp, loss))
        else:
            _isend(res.detach(), nxt)
            _isend(pending.detach(), nxt)
            saved.append((inp, (res, pending)))
        state["fwd"] += 1

    def do_bwd():
        inp, out = saved.popleft()
        if m.is_last:
            (out / microbatches).backward()
        else:
            g_res = _recv(shape, torch.bfloat16, nxt, group, dev)
            g_pending = _recv(shape, torch.bfloat16, nxt, group, dev)
            torch.autograd.backward(list(out), [g_res, g_pending])
        if not m.is_first:
            _isend(inp[0].grad, prv)
            _isend(inp[1].grad, prv)
        state["bwd"] += 1

    if schedule == "gpipe":
        for _ in range(microbatches):
            do_fwd()
        # fill-drain backward runs in reverse micro-batch order
        saved = deque(reversed(saved))
        for _ in range(microbatches):
            do_bwd()
    else:  # 1f1b
        warmup = min(m.world - 1 - m.stage, microbatches)
        for _ in range(warmup):
            do_fwd()
        while state["fwd"] < microbatches:
            do_fwd()
            do_bwd()
        while state["bwd"] < microbatches:
            do_bwd()
    for w in sends:
        w.wait()

    # everyone reports the same mean loss
    loss_val = torch.zeros(1, device=dev, dtype=torch.float32)
    if m.is_last:
        loss_val[0] = sum(float(l.detach()) for l in losses) / \
            microbatches
    dist.broadcast(loss_val, dist.get_global_rank(group, m.world - 1),
                   group=group)
    return float(loss_val)
