"""@torch_parallel(context_parallel=2) gang: the decorator must build the
dp x cp process-group grid and expose it via current.parallel."""

import torch

from metaflow_amd import FlowSpec, current, step, torch_parallel


class CpGridFlow(FlowSpec):
    @step
    def start(self):
        self.next(self.train, num_parallel=2)

    @torch_parallel(context_parallel=2)
    @step
    def train(self):
        import torch.distributed as dist

        p = current.parallel
        assert p.cp_degree == 2 and p.dp_degree == 1, p
        assert p.cp_rank == dist.get_rank()
        assert dist.get_world_size(p.cp_group) == 2
        assert dist.get_world_size(p.dp_group) == 1
        # a real collective over the cp group
        t = torch.tensor([float(dist.get_rank())])
        dist.all_reduce(t, group=p.cp_group)
        assert float(t) == 1.0, float(t)

        # ring attention over the decorator-provided group
        import math

        from metaflow_amd.parallel.ring_attention import ring_attention

        torch.manual_seed(0)
        B, H, S, D = 1, 2, 64, 128
        q = torch.randn(B, H, S, D)
        k = torch.randn(B, 1, S, D)
        v = torch.randn(B, 1, S, D)
        o = ring_attention(q, k, v, 1.0 / math.sqrt(D), group=p.cp_group)
        assert o.shape == (B, H, S, D)
        self.ok = True
        self.next(self.join)

    @step
    def join(self, inputs):
        assert all(i.ok for i in inputs)
        self.next(self.end)

    @step
    def end(self):
        pass


if __name__ == "__main__":
    CpGridFlow()
