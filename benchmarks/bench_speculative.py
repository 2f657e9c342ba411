#!/usr/bin/env python3
"""Speculative-decode throughput vs plain greedy generate (1 GPU).

    python benchmarks/bench_speculative.py [--target llama3-8b]
        [--draft tiny] [--k 4] [--prompt 128] [--new 128]

Written at the end of round 2 (GPU budget spent) — run on HW next
round; the CPU test suite (tests/test_speculative.py) already pins
token-exactness, so this script only measures speed + acceptance.
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
    p.add_argument("--target", default="llama3-8b",
                   choices=["llama3-8b", "tiny"])
    p.add_argument("--draft", default="tiny")
    p.add_argument("--k", type=int, default=4)
    p.add_argument("--prompt", type=int, default=128)
    p.add_argument("--new", type=int, default=128)
    args = p.parse_args()

    import torch

    from metaflow_amd.models.llama import LlamaConfig, LlamaForCausalLM
    from metaflow_amd.speculative import speculative_generate

    device = "cuda"
    torch.manual_seed(0)
    cfgs = {"llama3-8b": LlamaConfig.llama3_8b, "tiny": LlamaConfig.tiny}
    with torch.device(device):
        target = LlamaForCausalLM(cfgs[args.target]()).eval()
        draft = LlamaForCausalLM(cfgs[args.draft]()).eval()
    prompt = torch.randint(2, 1000, (1, args.prompt), device=device)

    torch.cuda.synchronize()
    t0 = time.time()
    ref = target.generate(prompt, args.new)
    torch.cuda.synchronize()
    greedy_s = time.time() - t0

    t0 = time.time()
    out, stats = speculative_generate(target, draft, prompt, args.new,
                                      k=args.k)
    torch.cuda.synchronize()
    spec_s = time.time() - t0
    exact = bool(torch.equal(out, ref))

    print(json.dumps({
        "metric": "speculative decode tokens/sec",
        "value": round(args.new / spec_s, 1), "unit": "tokens/s",
        "higher_is_better": True,
        "greedy_tokens_per_sec": round(args.new / greedy_s, 1),
        "speedup": round(greedy_s / spec_s, 3),
        "token_exact": exact,
        "acceptance": round(stats["accepted"] /
                            max(stats["proposed"], 1), 3),
        "target_steps": stats["target_steps"],
        "config": {"target": args.target, "draft": args.draft,
                   "k": args.k, "prompt": args.prompt,
                   "new": args.new},
    }), flush=True)


if __name__ == "__main__":
    main()
