"""Continuous-batching serving as a gang step: each rank pins one GPU,
loads the model, and drains its shard of the request stream through
metaflow_amd.serving.ContinuousBatcher (iteration-level scheduling over
the varlen flash-decode kernel).

    python examples/serving_flow.py run --num-requests 64
"""

from metaflow_amd import FlowSpec, Parameter, current, step, torch_parallel


class ServingFlow(FlowSpec):
    num_requests = Parameter("num_requests", default=32)
    ranks = Parameter("ranks", default=2)

    @step
    def start(self):
        import random

        random.seed(0)
        # synthetic request stream: random prompts, varied lengths
        self.requests = [
            ([random.randrange(2, 250) for _ in
              range(random.randrange(8, 64))],
             random.randrange(4, 16))
            for _ in range(int(self.num_requests))
        ]
        self.next(self.serve, num_parallel=int(self.ranks))

    @torch_parallel
    @step
    def serve(self):
        import time

        import torch

        from metaflow_amd.models.llama import LlamaConfig, LlamaForCausalLM
        from metaflow_amd.serving import ContinuousBatcher

        rank = current.parallel.node_index
        world = current.parallel.num_nodes
        shard = self.requests[rank::world]
        device = "cuda" if torch.cuda.is_available() else "cpu"
        torch.manual_seed(0)
        cfg = LlamaConfig.tiny(vocab=256, seq=512)
        model = LlamaForCausalLM(cfg).to(device).eval()
        batcher = ContinuousBatcher(model, max_batch=4, max_len=256)
        reqs = [batcher.submit(p, n) for p, n in shard]
        t0 = time.time()
        batcher.run()
        dt = time.time() - t0
        self.n_tokens = sum(len(r.generated) for r in reqs)
        self.tokens_per_sec = self.n_tokens / dt
        self.rank = rank
        self.next(self.join)

    @step
    def join(self, inputs):
        self.total_tokens = sum(i.n_tokens for i in inputs)
        self.throughput = sum(i.tokens_per_sec for i in inputs)
        self.next(self.end)

    @step
    def end(self):
        print("served %d tokens at %.0f tok/s aggregate"
              % (self.total_tokens, self.throughput))


if __name__ == "__main__":
    ServingFlow()
