"""Example: long-context Llama training with context parallelism.

    python examples/train_llama_cp.py run --num-gpus 8 --seq 65536

``@torch_parallel(context_parallel="all")`` makes the whole gang one
ring-attention group: the ranks share ONE batch with the sequence
sharded across them, and attention runs over the xGMI ring
(parallel/ring_attention.py) — so an 8-GPU node trains sequences 8x
longer than one GPU's HBM allows. Use an integer (e.g.
``context_parallel=4``) for a dp x cp grid instead; the flat-buffer
all-reduce over the whole gang averages cp shards exactly like extra
data-parallel ranks (see LlamaForCausalLM's docstring).
"""

from metaflow_amd import FlowSpec, Parameter, current, step, torch_parallel


class TrainLlamaCP(FlowSpec):
    num_gpus = Parameter("num_gpus", default=8, type=int)
    train_steps = Parameter("train_steps", default=20, type=int)
    batch = Parameter("batch", default=1, type=int)
    seq = Parameter("seq", default=65536, type=int)

    @step
    def start(self):
        self.next(self.train, num_parallel=self.num_gpus)

    @torch_parallel(context_parallel="all")
    @step
    def train(self):
        import torch
        import torch.distributed as dist

        from metaflow_amd.models.llama import LlamaConfig, LlamaForCausalLM
        from metaflow_amd.parallel.ddp import FlatParamModel, FusedAdamW

        p = current.parallel
        cp_degree = getattr(p, "cp_degree", 1)   # 1 when num_gpus == 1
        cp_group = getattr(p, "cp_group", None)
        cp_rank = getattr(p, "cp_rank", 0)
        dp_rank = getattr(p, "dp_rank", 0)
        use_gpu = torch.cuda.is_available()
        device = torch.device("cuda", 0) if use_gpu else torch.device("cpu")

        cfg = LlamaConfig.llama3_8b() if use_gpu else LlamaConfig.tiny()
        cfg.max_seq_len = max(cfg.max_seq_len, self.seq)
        seq = min(self.seq, cfg.max_seq_len)
        sc = seq // cp_degree             # this rank's sequence shard

        torch.manual_seed(1234)           # same init on every rank
        with torch.device(device):
            model = LlamaForCausalLM(cfg, cp_group=cp_group)
        flat = FlatParamModel(model)      # grad sync over the WHOLE gang
        flat.install_overlap_hooks()
        opt = FusedAdamW(flat, lr=3e-4)

        # one shared batch per cp ring; shard the sequence
        torch.manual_seed(99 + dp_rank)
        tok = torch.randint(0, cfg.vocab_size, (self.batch, seq + 1),
                            device=device)
        lo = cp_rank * sc
        inp = tok[:, lo:lo + sc].contiguous()
        tgt = tok[:, lo + 1:lo + 1 + sc].contiguous()

        for i in range(self.train_steps):
            flat.zero_grad()
            loss = model(inp, tgt)
            loss.backward()
            flat.finish_grad_sync()
            opt.step()
        self.final_loss = float(loss.detach())
        self.rank = dist.get_rank()
        self.next(self.join)

    @step
    def join(self, inputs):
        self.losses = sorted(round(i.final_loss, 3) for i in inputs)
        self.next(self.end)

    @step
    def end(self):
        print("per-rank final losses:", self.losses)


if __name__ == "__main__":
    TrainLlamaCP()
