import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from metaflow_amd.models.mixtral import MixtralConfig, MixtralForCausalLM

torch.manual_seed(0)
cfg = MixtralConfig.tiny(vocab=1024, seq=512)
with torch.device("cuda:0"):
    m = MixtralForCausalLM(cfg)
print("init ok", flush=True)
tok = torch.randint(0, cfg.vocab_size, (2, 513), device="cuda:0")
# stage 1: embedding + layers forward piecewise
x = m.embed(tok[:, :-1])
print("embed ok", flush=True)
for li, layer in enumerate(m.layers):
    res = x
    y = layer.input_norm(x)
    print("norm ok", li, flush=True); torch.cuda.synchronize()
    from metaflow_amd.ops import kernels as K
    B, S, _ = x.shape
    q = K.rope(layer.q_proj(y).view(B, S, cfg.num_heads, cfg.head_dim),
               m.cos_t, m.sin_t).transpose(1, 2)
    kk = K.rope(layer.k_proj(y).view(B, S, cfg.num_kv_heads, cfg.head_dim),
                m.cos_t, m.sin_t).transpose(1, 2)
    v = layer.v_proj(y).view(B, S, cfg.num_kv_heads,
                             cfg.head_dim).transpose(1, 2)
    torch.cuda.synchronize(); print("qkv+rope ok", li, flush=True)
    o = K.attention(q, kk, v)
    torch.cuda.synchronize(); print("attn fwd ok", li, flush=True)
    o = o.transpose(1, 2).reshape(B, S, cfg.num_heads * cfg.head_dim)
    x = res + layer.o_proj(o)
    moe_out = layer.moe(layer.post_norm(x))
    torch.cuda.synchronize(); print("moe ok", li, flush=True)
    x = x + moe_out
loss = m(tok[:, :-1], tok[:, 1:].contiguous())
torch.cuda.synchronize(); print("full fwd ok, loss", float(loss), flush=True)
loss.backward()
torch.cuda.synchronize(); print("bwd ok", flush=True)
