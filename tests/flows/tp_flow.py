"""@torch_parallel(tensor_parallel=2): TP Llama training end-to-end
through the gang scheduler on gloo."""

import torch

from metaflow_amd import FlowSpec, current, step, torch_parallel


class TPFlow(FlowSpec):
    @step
    def start(self):
        self.next(self.train, num_parallel=2)

    @torch_parallel(tensor_parallel=2)
    @step
    def train(self):
        import torch.distributed as dist

        from metaflow_amd.models.llama import LlamaConfig
        from metaflow_amd.models.llama_tp import TPLlamaForCausalLM
        from metaflow_amd.parallel.ddp import FlatParamModel, FusedAdamW

        p = current.parallel
        assert p.tp_degree == 2 and p.dp_degree == 1

        torch.manual_seed(5)
        cfg = LlamaConfig.tiny(vocab=256, seq=64)
        cfg.num_heads, cfg.num_kv_heads = 4, 2
        model = TPLlamaForCausalLM(cfg, p.tp_group)
        flat = FlatParamModel(model, bucket_mb=1, group=p.dp_group)
        flat.install_overlap_hooks()
        opt = FusedAdamW(flat, lr=1e-3)

        torch.manual_seed(9)      # same batch on both tp ranks
        tok = torch.randint(0, cfg.vocab_size, (1, 65))
        losses = []
        for _ in range(3):
            flat.zero_grad()
            loss = model(tok[:, :-1], tok[:, 1:].contiguous())
            loss.backward()
            flat.finish_grad_sync()
            opt.step()
            losses.append(float(loss.detach()))
        assert losses[-1] < losses[0], losses
        # both tp ranks must agree on the (replicated-activation) loss
        t = torch.tensor([losses[-1]])
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        assert abs(float(t) - losses[-1]) < 1e-3
        self.final_loss = losses[-1]
        self.rank = p.tp_rank
        self.next(self.join)

    @step
    def join(self, inputs):
        self.losses = [round(i.final_loss, 4) for i in inputs]
        self.next(self.end)

    @step
    def end(self):
        assert len(self.losses) == 2


if __name__ == "__main__":
    TPFlow()
