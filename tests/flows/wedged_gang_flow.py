"""Scheduler liveness: rank 1 wedges (sleeps forever, simulating a hung
RCCL rendezvous that no in-task @timeout can interrupt) on the first
attempt; MFX_TASK_STALL_TIMEOUT makes the runtime kill it, the gang
tears down and retries, and the retry completes."""

import os
import time

from metaflow_amd import FlowSpec, current, step, torch_parallel


class WedgedGangFlow(FlowSpec):
    @step
    def start(self):
        self.next(self.work, num_parallel=2)

    @torch_parallel
    @step
    def work(self):
        import torch.distributed as dist

        rank = current.parallel.node_index
        marker = os.path.join(os.environ["WEDGE_DIR"],
                              "attempted_%d" % rank)
        first = not os.path.exists(marker)
        with open(marker, "a") as f:
            f.write("x")
        if first and rank == 1:
            time.sleep(600)  # wedge: never reaches the collective
        import torch

        t = torch.tensor([float(rank)])
        dist.all_reduce(t)
        self.gang_sum = float(t)
        self.rank = rank
        self.next(self.join)

    @step
    def join(self, inputs):
        self.ranks = sorted(i.rank for i in inputs)
        self.next(self.end)

    @step
    def end(self):
        assert self.ranks == [0, 1], self.ranks


if __name__ == "__main__":
    WedgedGangFlow()
