"""Gang failure/recovery: rank 1 dies mid-gang on the first run
(GANG_FAIL=1); `resume` must rerun the WHOLE gang (gang steps clone
all-or-nothing) and complete."""

import os

from metaflow_amd import FlowSpec, current, step, torch_parallel


class GangFailFlow(FlowSpec):
    @step
    def start(self):
        self.next(self.work, num_parallel=2)

    @torch_parallel
    @step
    def work(self):
        import torch.distributed as dist

        rank = current.parallel.node_index
        if os.environ.get("GANG_FAIL") == "1" and rank == 1:
            os._exit(3)
        # gang ranks get disjoint CPU carve-outs (when cores allow)
        aff = os.environ.get("MFX_CPU_AFFINITY")
        if aff and hasattr(os, "sched_getaffinity"):
            cur = os.sched_getaffinity(0)
            lo, hi = (int(x) for x in aff.split("-"))
            assert cur == set(range(lo, hi + 1)), (aff, cur)
            self.affinity = aff
        # a collective proving the whole gang is alive
        import torch

        t = torch.tensor([float(rank)])
        dist.all_reduce(t)
        self.gang_sum = float(t)
        self.rank = rank
        self.next(self.join)

    @step
    def join(self, inputs):
        self.ranks = sorted(i.rank for i in inputs)
        self.gang_sum = inputs[0].gang_sum
        self.next(self.end)

    @step
    def end(self):
        assert self.ranks == [0, 1], self.ranks
        assert self.gang_sum == 1.0


if __name__ == "__main__":
    GangFailFlow()
