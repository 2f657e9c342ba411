#!/usr/bin/env python3
"""Continuous-batching serving throughput (ContinuousBatcher over the
varlen flash-decode kernel), hipGraph replay vs eager decode.

    python benchmarks/bench_serving.py [--model llama3-8b] [--slots 8]
        [--requests 24] [--prompt 256] [--new 64]
"""

import argparse
import json
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def run_one(model, graph, args, device):
    from metaflow_amd.serving import ContinuousBatcher

    batcher = ContinuousBatcher(model, max_batch=args.slots,
                                max_len=args.prompt + args.new + 8,
                                graph=graph,
                                prefill_chunk=args.prefill_chunk)
    random.seed(0)
    reqs = [batcher.submit(
        [random.randrange(2, 1000) for _ in
         range(random.randrange(args.prompt // 2, args.prompt))],
        args.new) for _ in range(args.requests)]
    import torch

    torch.cuda.synchronize()
    t0 = time.time()
    batcher.run()
    torch.cuda.synchronize()
    dt = time.time() - t0
    n = sum(len(r.generated) for r in reqs)
    return n, dt, batcher._graph is not None


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3-8b",
                   choices=["llama3-8b", "tiny"])
    p.add_argument("--slots", type=int, default=8)
    p.add_argument("--requests", type=int, default=24)
    p.add_argument("--prompt", type=int, default=256)
    p.add_argument("--new", type=int, default=64)
    p.add_argument("--prefill-chunk", type=int, default=None,
                   help="bound prompt tokens prefetched per step")
    args = p.parse_args()

    import torch

    from metaflow_amd.models.llama import LlamaConfig, LlamaForCausalLM

    device = "cuda"
    torch.manual_seed(0)
    cfg = (LlamaConfig.llama3_8b() if args.model == "llama3-8b"
           else LlamaConfig.tiny())
    with torch.device(device):
        model = LlamaForCausalLM(cfg).eval()

    for graph in (False, "auto"):
        n, dt, used_graph = run_one(model, graph, args, device)
        print(json.dumps({
            "metric": "serving tokens/sec", "value": round(n / dt, 1),
            "unit": "tokens/s", "higher_is_better": True,
            "graph_replay": used_graph,
            "config": {"model": args.model, "slots": args.slots,
                       "requests": args.requests,
                       "prompt_max": args.prompt, "new": args.new,
                       "prefill_chunk": args.prefill_chunk},
        }), flush=True)


if __name__ == "__main__":
    main()
