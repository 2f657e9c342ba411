"""@torch_parallel(pipeline_parallel=2): 2-stage pipeline Llama training
through the gang scheduler on gloo."""

import torch

from metaflow_amd import FlowSpec, current, step, torch_parallel


class PPFlow(FlowSpec):
    @step
    def start(self):
        self.next(self.train, num_parallel=2)

    @torch_parallel(pipeline_parallel="all")
    @step
    def train(self):
        from metaflow_amd.models.llama import LlamaConfig
        from metaflow_amd.models.llama_pp import (
            PPLlamaStage,
            pp_train_step,
        )

        p = current.parallel
        assert p.pp_degree == 2

        torch.manual_seed(4)
        cfg = LlamaConfig.tiny(vocab=256, seq=64)
        stage = PPLlamaStage(cfg, p.pp_group)
        opt = torch.optim.AdamW(stage.parameters(), lr=1e-3)

        torch.manual_seed(6)   # same batch on every stage
        tok = torch.randint(0, cfg.vocab_size, (2, 65))
        losses = []
        for _ in range(3):
            opt.zero_grad()
            loss = pp_train_step(stage, tok[:, :-1],
                                 tok[:, 1:].contiguous(),
                                 microbatches=2)
            opt.step()
            losses.append(loss)
        assert losses[-1] < losses[0], losses
        self.final_loss = losses[-1]
        self.stage = p.pp_rank
        self.next(self.join)

    @step
    def join(self, inputs):
        self.losses = sorted(round(i.final_loss, 4) for i in inputs)
        self.next(self.end)

    @step
    def end(self):
        # both stages report the same broadcast loss
        assert len(self.losses) == 2 and self.losses[0] == self.losses[1]


if __name__ == "__main__":
    PPFlow()
