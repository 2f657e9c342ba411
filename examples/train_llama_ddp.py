"""Example: data-parallel Llama training as a workflow.

    python examples/train_llama_ddp.py run --num-gpus 8 --train-steps 100

The @parallel gang step pins one rank per GPU; @torch_parallel brings up
RCCL; the flat-buffer DDP overlaps gradient all-reduce with backward; the
fused AdamW updates the whole parameter buffer in one kernel; @checkpoint
writes per-rank shards into the content-addressed store.
"""

from metaflow_amd import (
    FlowSpec,
    Parameter,
    card,
    checkpoint,
    current,
    step,
    torch_parallel,
)


class TrainLlamaDDP(FlowSpec):
    num_gpus = Parameter("num_gpus", default=8, type=int)
    train_steps = Parameter("train_steps", default=50, type=int)
    batch = Parameter("batch", default=8, type=int)
    seq = Parameter("seq", default=4096, type=int)
    model_size = Parameter("model_size", default="llama3-8b", type=str)

    @step
    def start(self):
        self.next(self.train, num_parallel=self.num_gpus)

    @card
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
        cfg = {"llama3-8b": LlamaConfig.llama3_8b,
               "llama3-70b": LlamaConfig.llama3_70b,
               "tiny": LlamaConfig.tiny}[str(self.model_size)]()

        torch.manual_seed(42)  # identical init on every rank
        with torch.device(device):
            model = LlamaForCausalLM(cfg)
        flat = FlatParamModel(model)
        flat.install_overlap_hooks()
        opt = FusedAdamW(flat, lr=3e-4)

        torch.manual_seed(1000 + rank)  # per-rank data
        seq = min(int(self.seq), cfg.max_seq_len)
        tokens = torch.randint(0, cfg.vocab_size,
                               (int(self.batch), seq + 1), device=device)
        losses = []
        for i in range(int(self.train_steps)):
            flat.zero_grad()
            loss = model(tokens[:, :-1], tokens[:, 1:].contiguous())
            loss.backward()
            flat.finish_grad_sync()
            opt.step()
            losses.append(float(loss.detach()))
        self.losses = losses
        self.rank = rank
        current.card.append("rank %d losses: %s"
                            % (rank, [round(l, 3) for l in losses[-5:]]),
                            title="Training")
        if rank == 0:
            current.checkpoint.save(opt.state_dict_tensors(), name="final")
        self.next(self.join)

    @step
    def join(self, inputs):
        self.final_losses = {i.rank: i.losses[-1] for i in inputs}
        self.next(self.end)

    @step
    def end(self):
        print("final losses per rank:", self.final_losses)


if __name__ == "__main__":
    TrainLlamaDDP()
