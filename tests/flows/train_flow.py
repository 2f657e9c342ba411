"""The FlowSpec-wrapped training path (BASELINE config 3 shape): an
@parallel gang step running the tiny Llama through the same flat-DDP +
fused-Adam machinery bench.py times. Used by GPU flow tests and as the
user-facing example."""

import os

from metaflow_amd import FlowSpec, Parameter, checkpoint, current, step
from metaflow_amd import torch_parallel


class TrainFlow(FlowSpec):
    num_nodes = Parameter("num_nodes", default=2, type=int)
    steps_n = Parameter("steps_n", default=4, type=int)

    @step
    def start(self):
        self.n = int(self.num_nodes)
        self.next(self.train, num_parallel=self.n)

    @checkpoint
    @torch_parallel
    @step
    def train(self):
        import torch

        from metaflow_amd.models.llama import LlamaConfig, LlamaForCausalLM
        from metaflow_amd.parallel.ddp import FlatParamModel, FusedAdamW

        rank = current.parallel.node_index
        device = torch.device("cuda", 0) if torch.cuda.is_available() \
            else torch.device("cpu")

        torch.manual_seed(42)  # same init on all ranks
        cfg = LlamaConfig.tiny(vocab=2048, seq=256)
        model = LlamaForCausalLM(cfg).to(device)
        flat = FlatParamModel(model, bucket_mb=1)
        flat.install_overlap_hooks()
        opt = FusedAdamW(flat, lr=1e-3)

        torch.manual_seed(1000 + rank)
        tokens = torch.randint(0, cfg.vocab_size, (1, 257), device=device)
        losses = []
        for _ in range(int(self.steps_n)):
            flat.zero_grad()
            loss = model(tokens[:, :-1], tokens[:, 1:].contiguous())
            loss.backward()
            flat.finish_grad_sync()
            opt.step()
            losses.append(float(loss.item()))
        self.losses = losses
        self.rank = rank
        # rank 0 checkpoints the (replicated) flat param shard
        if rank == 0:
            current.checkpoint.save(opt.state_dict_tensors(), name="final")
            self.ckpt_saved = True
        self.next(self.join)

    @step
    def join(self, inputs):
        ranks = sorted(i.rank for i in inputs)
        assert ranks == list(range(len(ranks))), ranks
        self.all_losses = {i.rank: i.losses for i in inputs}
        for r, ls in self.all_losses.items():
            assert ls[-1] == ls[-1], "NaN loss on rank %d" % r
        self.next(self.end)

    @step
    def end(self):
        self.ok = True


if __name__ == "__main__":
    TrainFlow()
