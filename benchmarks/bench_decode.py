#!/usr/bin/env python3
"""Decode (serving) throughput: KV-cache generate() tokens/sec.

    python benchmarks/bench_decode.py [--model llama3-8b] [--batch 8]
        [--prompt 512] [--new 128]

Measures prefill time and per-token decode rate separately. The decode
attention runs the split-K flash-decode kernel (ops/csrc/decode.hip)
straight over the cache allocation; prefill runs the flash fwd kernel.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3-8b",
                   choices=["llama3-8b", "tiny", "mixtral-tiny"])
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--prompt", type=int, default=512)
    p.add_argument("--new", type=int, default=128)
    args = p.parse_args()

    import torch

    from metaflow_amd.models.llama import LlamaConfig, LlamaForCausalLM
    from metaflow_amd.models.mixtral import (
        MixtralConfig,
        MixtralForCausalLM,
    )

    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda", 0) if use_gpu else torch.device("cpu")
    factories = {
        "llama3-8b": (LlamaConfig.llama3_8b, LlamaForCausalLM),
        "tiny": (LlamaConfig.tiny, LlamaForCausalLM),
        "mixtral-tiny": (MixtralConfig.tiny, MixtralForCausalLM),
    }
    cfg_fn, cls = factories[args.model]
    cfg = cfg_fn()
    prompt_len = min(args.prompt, cfg.max_seq_len - args.new)
    torch.manual_seed(0)
    with torch.device(device):
        model = cls(cfg)
    model.eval()
    tok = torch.randint(0, cfg.vocab_size, (args.batch, prompt_len),
                        device=device)

    # warmup (also JITs the cached path)
    model.generate(tok[:, :min(64, prompt_len)], 2)
    if use_gpu:
        torch.cuda.synchronize()

    from metaflow_amd.models.llama import KVCache

    t0 = time.time()
    cache = KVCache(cfg, args.batch, prompt_len + args.new, device,
                    dtype=model.embed.weight.dtype)
    with torch.no_grad():
        logits = model(tok, cache=cache)
    if use_gpu:
        torch.cuda.synchronize()
    prefill_s = time.time() - t0

    t0 = time.time()
    out = tok
    with torch.no_grad():
        for _ in range(args.new):
            nxt = logits[:, -1].float().argmax(-1, keepdim=True)
            out = torch.cat([out, nxt], dim=1)
            logits = model(nxt, cache=cache)
    if use_gpu:
        torch.cuda.synchronize()
    decode_s = time.time() - t0

    decode_tps = args.batch * args.new / decode_s
    print(json.dumps({
        "metric": "decode tokens/sec",
        "value": decode_tps,
        "unit": "tokens/s",
        "higher_is_better": True,
        "prefill_s": round(prefill_s, 3),
        "prefill_tps": round(args.batch * prompt_len / prefill_s, 1),
        "ms_per_decode_step": round(decode_s / args.new * 1000, 2),
        "config": {"model": args.model, "batch": args.batch,
                   "prompt": prompt_len, "new": args.new,
                   "device": str(device)},
    }), flush=True)


if __name__ == "__main__":
    main()
