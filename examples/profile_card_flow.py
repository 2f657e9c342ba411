"""Example: a perf card. @card(profile=True) reruns the step under
rocprofv3 (when the box has it) and the scheduler splices the
kernel-time breakdown into the task card; the card also carries GPU
telemetry sampled by the gpu_monitor sidecar.

    python examples/profile_card_flow.py run
    python -m metaflow_amd card ProfileCardFlow/<run>/train/<task>
"""

from metaflow_amd import FlowSpec, card, current, step


class ProfileCardFlow(FlowSpec):
    @step
    def start(self):
        self.next(self.train)

    @card(profile=True)
    @step
    def train(self):
        import torch

        from metaflow_amd.models.llama import LlamaConfig, LlamaForCausalLM

        device = "cuda" if torch.cuda.is_available() else "cpu"
        torch.manual_seed(0)
        cfg = LlamaConfig.tiny(vocab=512, seq=256)
        model = LlamaForCausalLM(cfg).to(device)
        toks = torch.randint(0, 512, (2, 257), device=device)
        for _ in range(3):
            loss = model(toks[:, :-1], toks[:, 1:].contiguous())
            loss.backward()
        self.loss = float(loss.detach())
        current.card.append("final loss %.3f" % self.loss,
                            title="Training")
        self.next(self.end)

    @step
    def end(self):
        pass


if __name__ == "__main__":
    ProfileCardFlow()
